from .quantization_config import (QuantizationConfig, QuantizationType,
                                  QuantizedDtype)
from .quantization_layers import QuantizedColumnParallel, QuantizedRowParallel
from .quantization_utils import quantize_symmetric, dequantize
from . import quantize
from . import microscaling
