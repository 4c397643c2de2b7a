from .topk import topk
from .argmax import argmax
