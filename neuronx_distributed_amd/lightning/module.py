"""NeuronLTModule (reference lightning/module.py:24-131): a LightningModule
base that defers model construction to ``initialize_parallel_model`` in
setup() and optimizer construction to ``initialize_parallel_optimizer`` in
configure_optimizers(); gradient clipping is a no-op (the NxD optimizer
clips, reference :89-95)."""

from typing import Any, Callable, Optional

import torch

try:
    import lightning.pytorch as pl

    _Base = pl.LightningModule
    _HAVE_LIGHTNING = True
except Exception:  # pragma: no cover
    _Base = object
    _HAVE_LIGHTNING = False


class NeuronLTModule(_Base):
    def __init__(self, nxd_config, model_fn: Callable,
                 optimizer_cls=torch.optim.AdamW,
                 optimizer_kwargs: Optional[dict] = None,
                 scheduler_cls=None, scheduler_kwargs: Optional[dict] = None):
        if not _HAVE_LIGHTNING:
            raise ImportError(
                "lightning is not installed; NeuronLTModule needs it")
        super().__init__()
        self.nxd_config = nxd_config
        self.model_fn = model_fn
        self.optimizer_cls = optimizer_cls
        self.optimizer_kwargs = optimizer_kwargs or {}
        self.scheduler_cls = scheduler_cls
        self.scheduler_kwargs = scheduler_kwargs or {}
        self.model = None
        self.automatic_optimization = True

    def setup(self, stage: str):
        from ..trainer import initialize_parallel_model

        if self.model is None:
            self.model = initialize_parallel_model(self.nxd_config,
                                                   self.model_fn)

    def forward(self, *args, **kwargs):
        return self.model(*args, **kwargs)

    def training_step(self, batch, batch_idx):
        loss = self.model(**batch)
        self.log("loss", loss, prog_bar=True)
        return loss

    def configure_optimizers(self):
        from ..trainer import initialize_parallel_optimizer

        opt = initialize_parallel_optimizer(
            self.nxd_config, self.optimizer_cls, self.model.parameters(),
            **self.optimizer_kwargs)
        if self.scheduler_cls is None:
            return opt
        sched = self.scheduler_cls(opt, **self.scheduler_kwargs)
        return [opt], [sched]

    def configure_gradient_clipping(self, *args, **kwargs):
        # clipping handled inside the NxD optimizer (reference :89-95)
        return
