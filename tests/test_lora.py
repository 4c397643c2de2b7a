"""LoRA: module swapping, zero-init equivalence, adapter training, merge."""

import torch

from dist_utils import run_distributed


def _lora_worker(rank, world):
    from neuronx_distributed_amd.parallel import parallel_state as ps
    import neuronx_distributed_amd.parallel as pl
    from neuronx_distributed_amd.models import get_config, LlamaForCausalLM
    from neuronx_distributed_amd.lora import LoraConfig, LoraModel

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    pl.model_parallel_manual_seed(0)
    torch.manual_seed(0)
    model = LlamaForCausalLM(get_config("tiny"))
    torch.manual_seed(3)
    x = torch.randint(0, 256, (2, 16))
    base_loss = model(x, labels=x).item()

    lora = LoraModel(model, LoraConfig(lora_rank=4))
    # B zero-init -> identical output
    loss = lora(x, labels=x)
    assert abs(loss.item() - base_loss) < 1e-5

    # only lora params trainable
    trainable = [n for n, p in lora.named_parameters() if p.requires_grad]
    assert trainable and all("lora_" in n for n in trainable)

    loss.backward()
    grads = [n for n, p in lora.named_parameters() if p.grad is not None
             and p.grad.abs().sum() > 0]
    # with B zero-init, dL/dA = 0 on the first step; B must get signal
    assert any("lora_B" in n for n in grads), grads

    # train a few steps: loss must move; then merge and match adapted output
    opt = torch.optim.AdamW([p for p in lora.parameters() if p.requires_grad],
                            lr=1e-2)
    for _ in range(3):
        opt.zero_grad()
        l = lora(x, labels=x)
        l.backward()
        opt.step()
    adapted = lora(x, labels=x).item()
    merged = lora.merge_lora()
    merged_loss = merged(x, labels=x).item()
    assert abs(adapted - merged_loss) < 1e-4, (adapted, merged_loss)
    return True


def test_lora_tp1():
    run_distributed(_lora_worker, world_size=1)


def test_lora_tp2():
    run_distributed(_lora_worker, world_size=2)


def _adapter_roundtrip_worker(rank, world):
    """Adapter state-dict roundtrip: train a few steps, save ONLY the
    adapter, load into a fresh wrap of the same base -> identical
    outputs; merged model matches too."""
    import copy

    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.lora import LoraConfig, LoraModel
    from neuronx_distributed_amd.models import get_config, LlamaForCausalLM

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    torch.manual_seed(0)
    base = LlamaForCausalLM(get_config("tiny"))
    base_copy = copy.deepcopy(base)

    cfg = LoraConfig(lora_rank=4, lora_alpha=8,
                     target_modules=["qkv_proj", "o_proj"])
    lm = LoraModel(base, cfg)
    opt = torch.optim.AdamW((p for p in lm.parameters() if p.requires_grad),
                            lr=1e-2)
    torch.manual_seed(1)
    x = torch.randint(0, 256, (2, 16))
    for _ in range(3):
        opt.zero_grad()
        loss = lm(x, labels=x)
        loss.backward()
        opt.step()
    ref_out = lm(x)
    sd = lm.get_adapter_state_dict()
    assert sd and all("lora_" in k for k in sd)

    lm2 = LoraModel(base_copy, cfg)
    lm2.load_adapter_state_dict(sd)
    out2 = lm2(x)
    assert torch.allclose(out2, ref_out, atol=1e-6)
    return float(ref_out.float().sum())


def test_lora_adapter_roundtrip():
    run_distributed(_adapter_roundtrip_worker, world_size=1)


def _trainer_lora_worker(rank, world):
    """lora_config flows through the trainer facade (phase 4): only
    adapter params are trainable and a step runs."""
    import neuronx_distributed_amd as nxd
    from neuronx_distributed_amd.lora import LoraConfig
    from neuronx_distributed_amd.models import get_config, LlamaForCausalLM

    cfg = nxd.neuronx_distributed_config(
        tensor_parallel_size=1,
        lora_config=LoraConfig(lora_rank=4, lora_alpha=8,
                               target_modules=["o_proj"]))
    model = nxd.initialize_parallel_model(
        cfg, lambda: LlamaForCausalLM(get_config("tiny")))
    trainable = [n for n, p in model.named_parameters() if p.requires_grad]
    assert trainable and all("lora_" in n for n in trainable), trainable[:4]
    opt = torch.optim.AdamW((p for p in model.parameters()
                             if p.requires_grad), lr=1e-2)
    x = torch.randint(0, 256, (2, 16))
    loss = model.run_train(x, labels=x)
    opt.step()
    assert torch.isfinite(loss)
    return float(loss.detach())


def test_trainer_lora_integration():
    run_distributed(_trainer_lora_worker, world_size=1)
