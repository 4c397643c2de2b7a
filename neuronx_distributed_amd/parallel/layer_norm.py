"""SP-aware LayerNorm (reference parallel_layers/layer_norm.py:7-40): tags
weight/bias with ``sequence_parallel_enabled`` so grads.py's SP all-reduce
finds them.  On GPU the forward dispatches to the fused HIP layernorm when
available."""

import torch
import torch.nn as nn


class LayerNorm(nn.LayerNorm):
    def __init__(self, *args, sequence_parallel_enabled: bool = False, **kwargs):
        super().__init__(*args, **kwargs)
        self.sequence_parallel_enabled = sequence_parallel_enabled
        if self.elementwise_affine:
            self.weight.sequence_parallel_enabled = sequence_parallel_enabled
            if self.bias is not None:
                self.bias.sequence_parallel_enabled = sequence_parallel_enabled
