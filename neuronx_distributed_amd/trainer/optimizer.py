"""NxDOptimizer: step orchestration (reference trainer/optimizer.py:122-147):
CP grad all-reduce -> SP layernorm grad all-reduce -> KV-shared grad sync ->
(zero1 | bucketed DP all-reduce + clip) -> inner step."""

import torch

from ..optimizer import NeuronZero1Optimizer
from ..parallel import grads as grads_mod, parallel_state as ps
from ..parallel.qkv_linear import allreduce_kv_shared_gradients


class NxDOptimizer(torch.optim.Optimizer):
    def __init__(self, optimizer, nxd_config):
        self.optimizer = optimizer
        self.nxd_config = nxd_config
        self._params = [
            p for g in optimizer.param_groups for p in g["params"]
        ]

    # delegate the torch.optim API
    @property
    def param_groups(self):
        return self.optimizer.param_groups

    @param_groups.setter
    def param_groups(self, v):
        self.optimizer.param_groups = v

    @property
    def state(self):
        return self.optimizer.state

    def state_dict(self):
        mp = self.nxd_config.get("mixed_precision_config") or {}
        if isinstance(self.optimizer, NeuronZero1Optimizer):
            return self.optimizer.state_dict(
                include_masters=mp.get("use_master_weights_in_ckpt", True))
        return self.optimizer.state_dict()

    def load_state_dict(self, sd):
        self.optimizer.load_state_dict(sd)

    def zero_grad(self, set_to_none: bool = False):
        self.optimizer.zero_grad(set_to_none=set_to_none)

    @property
    def grad_norm(self):
        return getattr(self.optimizer, "grad_norm", None)

    def _model_parameters(self):
        if isinstance(self.optimizer, NeuronZero1Optimizer):
            return [p for b in self.optimizer.buckets for p in b.params]
        return self._params

    def step(self, closure=None):
        params = self._model_parameters()
        grads_mod.allreduce_context_parallel_gradients(params)
        if self.nxd_config.get("sequence_parallel"):
            grads_mod.allreduce_sequence_parallel_gradients(params)
        allreduce_kv_shared_gradients(params)

        zero1 = isinstance(self.optimizer, NeuronZero1Optimizer)
        if not zero1:
            grads_mod.allreduce_gradients_for_parameters(params)
            opt_cfg = self.nxd_config["optimizer_config"]
            if opt_cfg.get("grad_clipping", True):
                grads_mod.clip_grad_norm(params,
                                         opt_cfg.get("max_grad_norm", 1.0))
        return self.optimizer.step(closure)
