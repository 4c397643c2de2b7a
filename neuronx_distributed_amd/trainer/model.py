"""NxDModel training wrapper (reference trainer/model.py:8-60)."""

import torch
import torch.nn as nn


class NxDModel(nn.Module):
    def __init__(self, module: nn.Module, nxd_config):
        super().__init__()
        self.module = module
        self.nxd_config = nxd_config
        self.pp_enabled = nxd_config["pipeline_parallel_size"] > 1

    def forward(self, *args, **kwargs):
        if self.pp_enabled:
            raise RuntimeError(
                "with pipeline parallelism call run_train()/run_eval() "
                "instead of forward() (reference trainer/model.py:49)")
        return self.module(*args, **kwargs)

    def run_train(self, *args, **kwargs):
        if self.pp_enabled:
            return self.module.run_train(*args, **kwargs)
        self.train()
        loss = self.module(*args, **kwargs)
        loss.backward()
        return loss

    def run_eval(self, *args, **kwargs):
        if self.pp_enabled:
            return self.module.run_eval(*args, **kwargs)
        self.eval()
        with torch.no_grad():
            return self.module(*args, **kwargs)

    @property
    def supports_tensor_position(self):
        return getattr(self.module, "supports_tensor_position", False) \
            and not self.pp_enabled

    @property
    def config(self):
        """Delegate to the wrapped model (generation/speculation utils
        read model.config)."""
        inner = self.module
        if self.pp_enabled:
            raise AttributeError("config is stage-local under PP")
        return inner.config

    def local_module(self):
        return self.module.local_module() if self.pp_enabled else self.module

    def local_named_parameters(self):
        if self.pp_enabled:
            return self.module.local_named_parameters()
        return self.module.named_parameters()

    def local_parameters(self):
        if self.pp_enabled:
            return self.module.local_parameters()
        return self.module.parameters()

    def state_dict(self, *args, **kwargs):
        return self.module.state_dict(*args, **kwargs)

    def load_state_dict(self, *args, **kwargs):
        return self.module.load_state_dict(*args, **kwargs)
