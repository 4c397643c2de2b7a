"""Lightning integration tests.

PyTorch Lightning is not installable in this image (no network), so these
tests install a MINIMAL STUB of the Lightning API surface our integration
touches (LightningModule base with ``log``/``automatic_optimization``,
``strategies.DDPStrategy`` with ``setup_distributed``) into sys.modules
BEFORE importing ``neuronx_distributed_amd.lightning``, then drive the
REAL NeuronLTStrategy / NeuronLTModule / NeuronCheckpointIO through the
documented PTL Trainer call sequence: setup_distributed -> setup("fit") ->
configure_optimizers -> training_step loop -> checkpoint IO.  Everything
exercised is OUR code; the stub only supplies the hook contract
(reference lightning/strategy.py:36,95-110, module.py:24-131)."""

import torch

from dist_utils import run_distributed


def _install_pl_stub():
    import sys
    import types

    pl = types.ModuleType("lightning")
    plpt = types.ModuleType("lightning.pytorch")
    strategies = types.ModuleType("lightning.pytorch.strategies")

    class LightningModule(torch.nn.Module):
        """Hook-contract subset: nn.Module + log() + trainer attr."""

        def __init__(self):
            super().__init__()
            self.trainer = None
            self.logged = {}

        def log(self, name, value, **kwargs):
            self.logged[name] = float(value)

    class DDPStrategy:
        def __init__(self, **kwargs):
            self._setup_dist_called = False

        def setup_distributed(self):
            # real PTL initializes torch.distributed here; the test harness
            # already did (dist_utils), matching an externally-launched job
            self._setup_dist_called = True

    plpt.LightningModule = LightningModule
    strategies.DDPStrategy = DDPStrategy
    pl.pytorch = plpt
    plpt.strategies = strategies
    sys.modules["lightning"] = pl
    sys.modules["lightning.pytorch"] = plpt
    sys.modules["lightning.pytorch.strategies"] = strategies


def _lt_e2e_worker(rank, world, tmpdir):
    _install_pl_stub()
    import neuronx_distributed_amd as nxd
    from neuronx_distributed_amd.lightning import (NeuronCheckpointIO,
                                                   NeuronLTModule,
                                                   NeuronLTStrategy)
    from neuronx_distributed_amd.models import LlamaForCausalLM, get_config
    from neuronx_distributed_amd.parallel import parallel_state as ps

    cfg = get_config("tiny")
    nxd_config = nxd.neuronx_distributed_config(
        tensor_parallel_size=world,
        optimizer_config={"zero_one_enabled": True, "grad_clipping": True,
                          "max_grad_norm": 1.0})

    # --- PTL trainer call sequence -----------------------------------
    strategy = NeuronLTStrategy(nxd_config=nxd_config)
    strategy.setup_distributed()
    assert strategy._setup_dist_called
    assert ps.model_parallel_is_initialized()
    assert ps.get_tensor_model_parallel_size() == world

    sk = strategy.distributed_sampler_kwargs
    assert sk == {"num_replicas": 1, "rank": 0}  # tp=world -> dp=1

    torch.manual_seed(0)
    module = NeuronLTModule(nxd_config,
                            model_fn=lambda: LlamaForCausalLM(cfg),
                            optimizer_kwargs={"lr": 1e-2})
    module.setup("fit")
    assert module.model is not None
    opt = module.configure_optimizers()

    torch.manual_seed(7)
    x = torch.randint(0, cfg.vocab_size, (2, 16))
    losses = []
    for step in range(4):
        loss = module.training_step({"input_ids": x, "labels": x}, step)
        loss.backward()
        module.configure_gradient_clipping(opt)  # must be a no-op
        opt.step()
        opt.zero_grad()
        losses.append(float(loss))
    assert losses[-1] < losses[0], losses
    assert "loss" in module.logged

    # --- checkpoint IO plugin ----------------------------------------
    io = NeuronCheckpointIO()
    state = {"state_dict": module.model.state_dict(), "step": 4}
    io.save_checkpoint(state, tmpdir)
    back = io.load_checkpoint(tmpdir)
    assert back["step"] == 4
    for k, v in module.model.state_dict().items():
        assert torch.equal(back["state_dict"][k], v), k
    return losses[-1]


def test_lightning_e2e_tp2():
    import tempfile

    with tempfile.TemporaryDirectory() as d:
        res = run_distributed(_lt_e2e_worker, world_size=2, args=(d,))
    assert abs(res[0] - res[1]) < 1e-5  # same loss on both tp ranks
