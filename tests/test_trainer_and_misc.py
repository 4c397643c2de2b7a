"""Trainer facade e2e (config -> model -> optimizer -> checkpoint ->
resume), GPT-2 TP=2 plumbing (BASELINE.json config #1), quantization,
timeline, serialization, tensor capture."""

import os
import tempfile

import pytest
import torch

from dist_utils import run_distributed


def _free_port():
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _trainer_worker(rank, world, tmpdir):
    import neuronx_distributed_amd as nxd
    from neuronx_distributed_amd.trainer import (
        neuronx_distributed_config, initialize_parallel_model,
        initialize_parallel_optimizer, save_checkpoint, load_checkpoint,
        has_checkpoint)
    from neuronx_distributed_amd.models import get_config, LlamaForCausalLM

    cfg = neuronx_distributed_config(
        tensor_parallel_size=world,
        optimizer_config={"zero_one_enabled": True, "grad_clipping": True,
                          "max_grad_norm": 1.0},
        activation_checkpoint_config="full")
    nxd.parallel.model_parallel_manual_seed(0)
    torch.manual_seed(0)
    model = initialize_parallel_model(cfg,
                                      lambda: LlamaForCausalLM(get_config("tiny")))
    opt = initialize_parallel_optimizer(cfg, torch.optim.AdamW,
                                        model.parameters(), lr=1e-3)
    torch.manual_seed(7)
    x = torch.randint(0, 256, (2, 16))
    losses = []
    for step in range(3):
        opt.zero_grad()
        loss = model(x, labels=x)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0]  # memorizing one batch must reduce loss
    assert opt.grad_norm is not None

    save_checkpoint(tmpdir, tag="3", model=model, optimizer=opt,
                    user_content={"step": 3}, num_kept=2)
    assert has_checkpoint(tmpdir)

    # fresh model/optimizer; resume; one more step must be identical to the
    # continuation of the original
    cont_loss = model(x, labels=x).item()

    torch.manual_seed(0)
    nxd.parallel.model_parallel_manual_seed(0)
    model2 = initialize_parallel_model(cfg,
                                       lambda: LlamaForCausalLM(get_config("tiny")))
    opt2 = initialize_parallel_optimizer(cfg, torch.optim.AdamW,
                                         model2.parameters(), lr=1e-3)
    uc = load_checkpoint(tmpdir, tag="latest_if_exists", model=model2,
                         optimizer=opt2)
    assert uc["step"] == 3
    resumed_loss = model2(x, labels=x).item()
    assert abs(resumed_loss - cont_loss) < 1e-5, (resumed_loss, cont_loss)
    return True


def test_trainer_e2e_checkpoint_resume():
    with tempfile.TemporaryDirectory() as d:
        run_distributed(_trainer_worker, world_size=2, args=(d,))


def _gpt2_worker(rank, world):
    """BASELINE.json config #1: 2-layer GPT-2-small, TP=2, CPU plumbing."""
    from neuronx_distributed_amd.parallel import parallel_state as ps
    import neuronx_distributed_amd.parallel as pl
    from neuronx_distributed_amd.models.gpt2 import (get_gpt2_config,
                                                     GPT2LMHeadModel)

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    pl.model_parallel_manual_seed(0)
    torch.manual_seed(0)
    model = GPT2LMHeadModel(get_gpt2_config("gpt2-small-2l"))
    torch.manual_seed(1)
    x = torch.randint(0, 50304, (2, 32))
    loss = model(x, labels=x)
    loss.backward()
    assert torch.isfinite(loss)
    return loss.item()


def test_gpt2_small_tp2_plumbing():
    outs = run_distributed(_gpt2_worker, world_size=2)
    assert abs(outs[0] - outs[1]) < 1e-6
    # vs tp1 golden
    ref = run_distributed(_gpt2_worker, world_size=1)[0]
    assert abs(outs[0] - ref) < 5e-2, (outs[0], ref)


def _quant_worker(rank, world):
    from neuronx_distributed_amd.parallel import parallel_state as ps
    import neuronx_distributed_amd.parallel as pl
    from neuronx_distributed_amd.models import get_config, LlamaForCausalLM
    from neuronx_distributed_amd.quantization import quantize, QuantizationConfig

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    pl.model_parallel_manual_seed(0)
    torch.manual_seed(0)
    model = LlamaForCausalLM(get_config("tiny")).eval()
    torch.manual_seed(1)
    x = torch.randint(0, 256, (1, 16))
    with torch.no_grad():
        ref_logits = model(x)
        quantize.convert(model, QuantizationConfig())
        q_logits = model(x)
    rel = (q_logits - ref_logits).abs().max() / ref_logits.abs().max()
    assert rel < 0.1, rel.item()  # int8 per-channel ~ close
    return True


def test_quantized_layers_tp2():
    run_distributed(_quant_worker, world_size=2)


def test_serialization_roundtrip():
    from neuronx_distributed_amd.utils.serialization import SerializationManager

    sm = SerializationManager()
    obj = {"a": torch.randn(3), "b": [1, (torch.ones(2), "x")], "c": None}
    skel, metas, tensors = sm.serialize(obj)
    assert len(tensors) == 2 and metas[0].shape == (3,)
    out = sm.deserialize(skel, tensors)
    assert torch.equal(out["a"], obj["a"])
    assert out["b"][1][1] == "x"


def test_timeline(tmp_path):
    from neuronx_distributed_amd.utils.timeline import Timeline

    p = str(tmp_path / "trace.json")
    tl = Timeline(p, rank=0)
    tl.mark_event_start("fwd")
    tl.mark_event_end("fwd")
    tl.mark_step_end(gather=False)
    import json

    data = json.load(open(p))
    assert data["traceEvents"][0]["name"] == "fwd"


def test_tensor_capture():
    from neuronx_distributed_amd.utils.tensor_capture import (
        enable_tensor_capture, get_captured_tensors, disable_tensor_capture)

    m = torch.nn.Sequential(torch.nn.Linear(4, 4), torch.nn.ReLU())
    enable_tensor_capture(m, ["0"])
    m(torch.randn(2, 4))
    cap = get_captured_tensors()
    assert "0" in cap and cap["0"].shape == (2, 4)
    disable_tensor_capture()


def test_checkpoint_converter_roundtrip():
    from neuronx_distributed_amd.scripts import CheckpointConverterBase

    torch.manual_seed(0)
    full = {
        "model.layers.0.self_attn.q_proj.weight": torch.randn(16, 8),
        "model.layers.0.self_attn.o_proj.weight": torch.randn(8, 16),
        "model.layers.0.mlp.gate_up_proj.weight": torch.randn(32, 8),
        "model.norm.weight": torch.randn(8),
    }
    c = CheckpointConverterBase()
    shards = c.shard_full_checkpoint(full, tp_degree=2)
    assert shards[0]["model.layers.0.self_attn.q_proj.weight"].shape == (8, 8)
    assert shards[0]["model.layers.0.self_attn.o_proj.weight"].shape == (8, 8)
    assert shards[0]["model.layers.0.mlp.gate_up_proj.weight"].shape == (16, 8)
    assert shards[0]["model.norm.weight"].shape == (8,)
    merged = c.merge_sharded_checkpoints(shards)
    for k in full:
        assert torch.equal(merged[k], full[k]), k


def test_top_level_api_surface():
    import neuronx_distributed_amd as nxd

    for attr in ("parallel_layers", "pipeline", "kernels", "utils",
                 "ModelBuilder", "NxDModel", "shard_checkpoint",
                 "NxDParallelState", "neuronx_distributed_config",
                 "initialize_parallel_model", "initialize_parallel_optimizer",
                 "save_checkpoint", "load_checkpoint"):
        assert hasattr(nxd, attr), attr
    from neuronx_distributed_amd.parallel_layers import (
        ColumnParallelLinear, RowParallelLinear, ParallelEmbedding,
        parallel_cross_entropy, initialize_model_parallel, clip_grad_norm,
        PARALLEL_MODULES, PARALLEL_FUNCTIONS)


def test_training_metrics(tmp_path):
    import time
    from neuronx_distributed_amd.utils.training_metrics import (
        Metric, Throughput, TrainingMetrics)

    tp = Throughput(batch_size=4, world_size=2, grad_accum_usteps=2,
                    moving_avg_window_size=4)
    time.sleep(0.01)
    v1 = tp.get_throughput()
    assert v1 > 0
    for _ in range(4):
        time.sleep(0.002)
        v = tp.get_throughput()
    assert v > 0

    f = str(tmp_path / "results.json")
    tm = TrainingMetrics(f)
    tm.store_parameters({"Model": "llama2-7b", "World size": 2})
    tm.store_metrics([Metric("Throughput", 123.4, "seq/s"),
                      Metric("FinalLoss", 2.5)])
    tm.store_metrics([Metric("Throughput", 125.0, "seq/s")])
    import json
    d = json.load(open(f))
    assert d["parameters"]["Model"] == "llama2-7b"
    assert len(d["metrics"]) == 3
    assert d["metrics"][0]["MetricName"] == "Throughput"


def _neox_worker(rank, world):
    """GPT-NeoX (parallel residual, partial rotary, LN+bias) trains at TP=world
    and TP2 matches TP1."""
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.models import (GPTNeoXForCausalLM,
                                                get_neox_config)

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(0)
    m = GPTNeoXForCausalLM(get_neox_config("gpt-neox-tiny"))
    torch.manual_seed(1)
    x = torch.randint(0, 256, (2, 16))
    loss = m(x, labels=x)
    loss.backward()
    assert all(p.grad is not None for p in m.parameters())
    return float(loss.detach())


def test_gpt_neox_tp2_matches_tp1():
    from tests.dist_utils import run_distributed
    tp1 = run_distributed(_neox_worker, world_size=1)[0]
    tp2 = run_distributed(_neox_worker, world_size=2)
    assert abs(tp2[0] - tp2[1]) < 1e-5
    assert abs(tp1 - tp2[0]) < 5e-3, (tp1, tp2)


def test_bench_multirank_cpu_smoke():
    """bench.py's multi-rank path (the driver's N>1 launch shape) runs end
    to end on gloo with the tiny model, prints the contract JSON line, and
    enables SP by default at TP>1."""
    import json
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", str(_free_port()), os.path.join(repo, "bench.py"),
           "--gpus", "2", "--steps", "1", "--warmup", "0", "--model", "tiny",
           "--seq", "64", "--batch", "2", "--microbatch", "1"]
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=420,
                         cwd=repo)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    r = json.loads(line)
    assert r["n_gpus"] == 2 and r["unit"] == "tokens/s"
    assert r["config"]["parallelism"] == "tp2_sp"
    assert r["value"] > 0


def test_bench_pp_cpu_smoke():
    """bench.py --pp 2: the 1F1B pipeline path runs end to end (tiny model,
    gloo) and reports the contract line with parallelism tp1_pp2."""
    import json
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", str(_free_port()), os.path.join(repo, "bench.py"),
           "--gpus", "2", "--steps", "1", "--warmup", "0", "--model", "tiny",
           "--seq", "64", "--batch", "4", "--microbatch", "1",
           "--tp", "1", "--pp", "2"]
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=420,
                         cwd=repo)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    r = json.loads(line)
    assert r["config"]["parallelism"] == "tp1_pp2"
    assert r["value"] > 0


def test_bench_tp_pp_3d_cpu_smoke():
    """bench.py --tp 2 --pp 2 on world 4 (gloo): the combined TP+PP path
    (BASELINE config #3 shape) runs end to end."""
    import json
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "4", "--master-addr", "127.0.0.1",
           "--master-port", str(_free_port()), os.path.join(repo, "bench.py"),
           "--gpus", "4", "--steps", "1", "--warmup", "0", "--model", "tiny",
           "--seq", "64", "--batch", "4", "--microbatch", "1",
           "--tp", "2", "--pp", "2"]
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=420,
                         cwd=repo)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    r = json.loads(line)
    assert r["config"]["parallelism"] == "tp2_pp2"
    assert r["value"] > 0


def test_bench_tp4_cpu_smoke():
    """bench.py at TP=4 (world 4, gloo): exercises KV-head replication
    (tiny model: 2 kv heads x tp4 -> kv_size_multiplier 2) through the
    full training step."""
    import json
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "4", "--master-addr", "127.0.0.1",
           "--master-port", str(_free_port()), os.path.join(repo, "bench.py"),
           "--gpus", "4", "--steps", "1", "--warmup", "0", "--model", "tiny",
           "--seq", "64", "--batch", "4", "--microbatch", "2"]
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=420,
                         cwd=repo)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    r = json.loads(line)
    assert r["config"]["parallelism"] == "tp4_sp"
    assert r["value"] > 0


def _dp_loader_worker(rank, world):
    """create_dp_dataloader: TP ranks share batches; DP shards them."""
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.utils.batch_utils import create_dp_dataloader

    # tp=2 at world 2 -> dp=1: both ranks see identical batches
    ps.initialize_model_parallel(tensor_model_parallel_size=2)
    data = torch.arange(32).reshape(16, 2)
    dl = create_dp_dataloader(list(data), batch_size=4, shuffle=True, seed=1)
    first = next(iter(dl))
    return [int(x) for x in first.reshape(-1)]


def _dp_loader_worker_dp2(rank, world):
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.utils.batch_utils import create_dp_dataloader

    ps.initialize_model_parallel(tensor_model_parallel_size=1)  # dp=2
    data = list(range(16))
    dl = create_dp_dataloader(data, batch_size=2, shuffle=False)
    seen = [int(x) for b in dl for x in b]
    return seen


def test_dp_dataloader():
    out = run_distributed(_dp_loader_worker, world_size=2)
    assert out[0] == out[1]  # same tp replica -> same data
    shards = run_distributed(_dp_loader_worker_dp2, world_size=2)
    assert set(shards[0]).isdisjoint(shards[1])
    assert len(shards[0]) == len(shards[1]) == 8


def _trainer_pp_worker(rank, world):
    """High-level trainer facade at PP=2: neuronx_distributed_config ->
    initialize_parallel_model (pipeline wrap) -> parallel optimizer ->
    run_train steps reduce the loss."""
    import neuronx_distributed_amd as nxd
    from neuronx_distributed_amd.models import get_config, LlamaForCausalLM
    from neuronx_distributed_amd.models.llama import LlamaDecoderLayer

    cfg = nxd.neuronx_distributed_config(
        tensor_parallel_size=1, pipeline_parallel_size=world,
        pipeline_config={"transformer_layer_cls": LlamaDecoderLayer,
                         "num_microbatches": 2,
                         "input_names": ["input_ids", "labels"]},
        optimizer_config={"zero_one_enabled": True, "grad_clipping": True,
                          "max_grad_norm": 1.0})
    model = nxd.initialize_parallel_model(
        cfg, lambda: LlamaForCausalLM(get_config("tiny")))
    opt = nxd.initialize_parallel_optimizer(cfg, torch.optim.AdamW,
                                            model.parameters(), lr=1e-2)
    torch.manual_seed(1)
    x = torch.randint(0, 256, (4, 16))
    losses = []
    for _ in range(4):
        opt.zero_grad()
        loss = model.run_train(input_ids=x, labels=x)
        opt.step()
        losses.append(float(loss))
    assert losses[-1] < losses[0], losses
    return losses[-1]


def test_trainer_facade_pp2():
    out = run_distributed(_trainer_pp_worker, world_size=2)
    assert abs(out[0] - out[1]) < 1e-5


def _neox_generate_worker(rank, world):
    """GPT-NeoX KV-cache generation (partial rotary + cache) matches full
    re-forward greedy decoding."""
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.models import (GPTNeoXForCausalLM,
                                                get_neox_config)
    from neuronx_distributed_amd.inference.generation import generate

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    torch.manual_seed(0)
    m = GPTNeoXForCausalLM(get_neox_config("gpt-neox-tiny")).eval()
    torch.manual_seed(1)
    x = torch.randint(0, 256, (2, 9))
    out = generate(m, x, max_new_tokens=6)
    seq = x
    for _ in range(6):
        logits = m(seq)
        seq = torch.cat([seq, logits[:, -1, :].argmax(-1, keepdim=True)], 1)
    assert torch.equal(out, seq), (out, seq)
    return 0.0


def test_neox_generate():
    run_distributed(_neox_generate_worker, world_size=1)


def _gpt2_generate_worker(rank, world):
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.models.gpt2 import (GPT2LMHeadModel,
                                                     get_gpt2_config)
    from neuronx_distributed_amd.inference.generation import generate

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    torch.manual_seed(0)
    m = GPT2LMHeadModel(get_gpt2_config("gpt2-tiny")).eval()
    torch.manual_seed(1)
    x = torch.randint(0, 256, (2, 9))
    out = generate(m, x, max_new_tokens=6)
    seq = x
    for _ in range(6):
        logits = m(seq)
        seq = torch.cat([seq, logits[:, -1, :].argmax(-1, keepdim=True)], 1)
    assert torch.equal(out, seq), (out, seq)
    return 0.0


def test_gpt2_generate():
    run_distributed(_gpt2_generate_worker, world_size=1)


def _lr_sched_worker(rank, world):
    """torch LR schedulers drive NxDOptimizer/zero1 param groups."""
    import neuronx_distributed_amd as nxd
    from neuronx_distributed_amd.models import get_config, LlamaForCausalLM

    cfg = nxd.neuronx_distributed_config(tensor_parallel_size=1)
    model = nxd.initialize_parallel_model(
        cfg, lambda: LlamaForCausalLM(get_config("tiny")))
    opt = nxd.initialize_parallel_optimizer(cfg, torch.optim.AdamW,
                                            model.parameters(), lr=1e-2)
    sched = torch.optim.lr_scheduler.LambdaLR(
        opt, lr_lambda=lambda step: 1.0 / (1 + step))
    x = torch.randint(0, 256, (2, 16))
    lrs = []
    for _ in range(3):
        opt.zero_grad()
        model(x, labels=x).backward()
        opt.step()
        sched.step()
        lrs.append(opt.param_groups[0]["lr"])
    assert abs(lrs[0] - 5e-3) < 1e-9 and abs(lrs[1] - 1e-2 / 3) < 1e-9, lrs
    return lrs[-1]


def test_lr_scheduler_integration():
    run_distributed(_lr_sched_worker, world_size=1)


def _meta_init_worker(rank, world):
    """meta_device_init: model builds on meta (no host RAM for weights)
    and materializes through the trainer phases; a train step runs."""
    import neuronx_distributed_amd as nxd
    from neuronx_distributed_amd.models import get_config, LlamaForCausalLM

    cfg = nxd.neuronx_distributed_config(
        tensor_parallel_size=1,
        model_init_config={"meta_device_init": True})
    model = nxd.initialize_parallel_model(
        cfg, lambda: LlamaForCausalLM(get_config("tiny")))
    for p in model.parameters():
        assert p.device.type != "meta"
    opt = nxd.initialize_parallel_optimizer(cfg, torch.optim.AdamW,
                                            model.parameters(), lr=1e-2)
    x = torch.randint(0, 256, (2, 16))
    loss = model(x, labels=x)
    loss.backward()
    opt.step()
    assert torch.isfinite(loss)
    return float(loss.detach())


def test_meta_device_init():
    run_distributed(_meta_init_worker, world_size=1)


def _native_converter_worker(rank, world):
    """Converter roundtrips a NATIVE llama state dict (qkv weight_q/k/v
    names): shard tp2 -> merge -> exact equality; and the tp2 shard
    matches what a tp2-built model holds."""
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.models import get_config, LlamaForCausalLM
    from neuronx_distributed_amd.scripts.checkpoint_converter import (
        CheckpointConverterBase)

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    torch.manual_seed(0)
    dense = LlamaForCausalLM(get_config("tiny"))
    full = {k: v for k, v in dense.state_dict().items()}
    conv = CheckpointConverterBase()
    shards = conv.shard_full_checkpoint(full, 2)
    back = conv.merge_sharded_checkpoints(shards)
    for k in full:
        assert torch.equal(full[k], back[k]), k
    # sharded shapes actually shrink on the parallel dims
    assert shards[0]["lm_head.weight"].shape[0] * 2 == \
        full["lm_head.weight"].shape[0]
    assert shards[0][
        "model.layers.0.self_attn.qkv_proj.weight_q"].shape[0] * 2 == \
        full["model.layers.0.self_attn.qkv_proj.weight_q"].shape[0]
    return 0.0


def test_native_converter_roundtrip():
    run_distributed(_native_converter_worker, world_size=1)


def _hf_convert_worker(rank, world):
    """HF llama checkpoint names convert to the native layout, load into
    the model, and the fused gate_up halves land in the right place."""
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.models import get_config, LlamaForCausalLM
    from neuronx_distributed_amd.scripts.checkpoint_converter import (
        convert_hf_llama_state_dict)

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    cfg = get_config("tiny")
    torch.manual_seed(0)
    m = LlamaForCausalLM(cfg)
    native = m.state_dict()

    # build an HF-shaped dict from the native one (inverse mapping)
    hf = {}
    for k, v in native.items():
        if k.endswith("qkv_proj.weight_q"):
            hf[k.replace("qkv_proj.weight_q", "q_proj.weight")] = v
        elif k.endswith("qkv_proj.weight_k"):
            hf[k.replace("qkv_proj.weight_k", "k_proj.weight")] = v
        elif k.endswith("qkv_proj.weight_v"):
            hf[k.replace("qkv_proj.weight_v", "v_proj.weight")] = v
        elif k.endswith("gate_up_proj.weight"):
            I = v.shape[0] // 2
            hf[k.replace("gate_up_proj.weight", "gate_proj.weight")] = v[:I]
            hf[k.replace("gate_up_proj.weight", "up_proj.weight")] = v[I:]
        else:
            hf[k] = v
    hf["model.layers.0.self_attn.rotary_emb.inv_freq"] = torch.zeros(4)

    back = convert_hf_llama_state_dict(hf)
    torch.manual_seed(1)
    m2 = LlamaForCausalLM(cfg)
    m2.load_state_dict(back)
    x = torch.randint(0, 256, (2, 8))
    assert torch.allclose(m(x), m2(x), atol=1e-6)
    return 0.0


def test_hf_llama_conversion():
    run_distributed(_hf_convert_worker, world_size=1)


def _hf_mixtral_worker(rank, world):
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.models import (MixtralForCausalLM,
                                                get_moe_config)
    from neuronx_distributed_amd.scripts.checkpoint_converter import (
        convert_hf_mixtral_state_dict)

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    cfg = get_moe_config("tiny-moe")
    torch.manual_seed(0)
    m = MixtralForCausalLM(cfg)
    native = m.state_dict()

    hf = {}
    for k, v in native.items():
        if k.endswith("qkv_proj.weight_q"):
            hf[k.replace("qkv_proj.weight_q", "q_proj.weight")] = v
        elif k.endswith("qkv_proj.weight_k"):
            hf[k.replace("qkv_proj.weight_k", "k_proj.weight")] = v
        elif k.endswith("qkv_proj.weight_v"):
            hf[k.replace("qkv_proj.weight_v", "v_proj.weight")] = v
        elif k.endswith("router.linear_router.weight"):
            hf[k.replace("router.linear_router.weight", "gate.weight")] = v
        elif k.endswith("expert_mlps.gate_up_proj.weight"):
            base = k.rsplit(".expert_mlps.gate_up_proj.weight", 1)[0]
            I = v.shape[2] // 2
            for e in range(v.shape[0]):
                hf[f"{base}.experts.{e}.w1.weight"] = v[e, :, :I].t().contiguous()
                hf[f"{base}.experts.{e}.w3.weight"] = v[e, :, I:].t().contiguous()
        elif k.endswith("expert_mlps.down_proj.weight"):
            base = k.rsplit(".expert_mlps.down_proj.weight", 1)[0]
            for e in range(v.shape[0]):
                hf[f"{base}.experts.{e}.w2.weight"] = v[e].t().contiguous()
        else:
            hf[k] = v

    back = convert_hf_mixtral_state_dict(hf, cfg.num_local_experts)
    torch.manual_seed(1)
    m2 = MixtralForCausalLM(cfg)
    m2.load_state_dict(back)
    x = torch.randint(0, 256, (2, 8))
    with torch.no_grad():
        assert torch.allclose(m(x), m2(x), atol=1e-6)
    return 0.0


def test_hf_mixtral_conversion():
    run_distributed(_hf_mixtral_worker, world_size=1)


def test_serving_endpoint():
    """The FastAPI serving demo answers /generate in-process."""
    import importlib.util

    spec = importlib.util.spec_from_file_location(
        "serve_demo", os.path.join(os.path.dirname(os.path.dirname(
            os.path.abspath(__file__))), "examples", "inference", "serve.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    from starlette.testclient import TestClient

    saved_env = {k: os.environ.get(k)
                 for k in ("NXDA_FAST_INIT", "MASTER_ADDR", "MASTER_PORT")}
    try:
        app = mod.build_app("tiny")
        client = TestClient(app)
        r = client.get("/health")
        assert r.status_code == 200 and r.json()["status"] == "ok"
        r = client.post("/generate", json={"token_ids": [[1, 2, 3]],
                                           "max_new_tokens": 4})
        assert r.status_code == 200
        ids = r.json()["token_ids"]
        assert len(ids) == 1 and len(ids[0]) == 4
    finally:
        # leave the pytest process clean for later tests
        import torch.distributed as dist
        from neuronx_distributed_amd.parallel import parallel_state as ps
        if ps.model_parallel_is_initialized():
            ps.destroy_model_parallel()
        if dist.is_initialized():
            dist.destroy_process_group()
        for k, v in saved_env.items():
            if v is None:
                os.environ.pop(k, None)
            else:
                os.environ[k] = v


def _pph_worker(rank, world):
    import neuronx_distributed_amd as nxd
    from neuronx_distributed_amd.models import get_config, LlamaForCausalLM
    from neuronx_distributed_amd.trainer.post_partition_hooks import (
        clear_post_partition_hooks, register_post_partition_hook)

    calls = []

    @register_post_partition_hook
    def mark(model, cfg):
        calls.append(type(model).__name__)

    try:
        cfg = nxd.neuronx_distributed_config(tensor_parallel_size=1)
        nxd.initialize_parallel_model(
            cfg, lambda: LlamaForCausalLM(get_config("tiny")))
    finally:
        clear_post_partition_hooks()
    assert calls == ["LlamaForCausalLM"], calls
    return 0.0


def test_post_partition_hooks():
    run_distributed(_pph_worker, world_size=1)


def test_checkpoint_storage_backend(tmp_path):
    import pytest as _pytest
    from neuronx_distributed_amd.trainer.checkpoint_storage import (
        LocalStorage, get_storage)

    st = get_storage(str(tmp_path))
    assert isinstance(st, LocalStorage)
    st.save_object({"a": torch.ones(3)}, "tag/model/x.pt")
    assert st.exists("tag/model/x.pt")
    obj = st.load_object("tag/model/x.pt")
    assert torch.equal(obj["a"], torch.ones(3))
    st.write_text("tag/done", "done")
    assert "tag" in st.listdir()
    st.remove_tree("tag")
    assert not st.exists("tag/done")
    # s3:// now routes through FsspecStorage; without s3fs installed the
    # protocol resolution fails loudly (with s3fs on a cluster it works)
    with _pytest.raises((ImportError, ValueError)):
        get_storage("s3://bucket/prefix")


def _nxdmodel_generate_worker(rank, world):
    """generate() works directly on the trainer's NxDModel wrapper."""
    import neuronx_distributed_amd as nxd
    from neuronx_distributed_amd.models import get_config, LlamaForCausalLM

    cfg = nxd.neuronx_distributed_config(tensor_parallel_size=1)
    model = nxd.initialize_parallel_model(
        cfg, lambda: LlamaForCausalLM(get_config("tiny")))
    model.eval()
    x = torch.randint(0, 256, (1, 8))
    out = nxd.generate(model, x, max_new_tokens=4)
    assert out.shape == (1, 12)
    return 0.0


def test_nxdmodel_generate():
    run_distributed(_nxdmodel_generate_worker, world_size=1)
