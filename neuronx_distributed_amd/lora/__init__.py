from .config import LoraConfig
from .model import LoraModel
from .layer import LoraLinear, LoraEmbedding
from .tp_layer import LoraParallelLinear
