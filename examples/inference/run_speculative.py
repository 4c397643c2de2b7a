#!/usr/bin/env python3
"""Speculative decoding demo: draft-and-verify (a small draft model) and
Medusa heads, both exactly matching target-only greedy decoding.

  python examples/inference/run_speculative.py            # CPU demo
  torchrun --nproc-per-node 1 examples/inference/run_speculative.py
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))

import torch
import torch.distributed as dist


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="test-d128")
    p.add_argument("--draft-layers", type=int, default=1)
    p.add_argument("--batch", type=int, default=2)
    p.add_argument("--prompt-len", type=int, default=64)
    p.add_argument("--new-tokens", type=int, default=64)
    p.add_argument("--spec-len", type=int, default=4)
    args = p.parse_args()

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29791")
    os.environ.setdefault("NXDA_FAST_INIT", "1")
    on_gpu = torch.cuda.is_available()
    dist.init_process_group("nccl" if on_gpu else "gloo",
                            rank=int(os.environ.get("RANK", "0")),
                            world_size=int(os.environ.get("WORLD_SIZE", "1")))

    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.models import get_config, LlamaForCausalLM
    from neuronx_distributed_amd.inference.generation import generate
    from neuronx_distributed_amd.inference.speculation import (
        medusa_generate, speculative_generate)
    from neuronx_distributed_amd.utils.medusa_utils import MedusaHead

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    device = torch.device("cuda") if on_gpu else torch.device("cpu")
    dtype = torch.bfloat16 if on_gpu else torch.float32
    prev = torch.get_default_dtype()
    torch.set_default_dtype(dtype)
    torch.manual_seed(0)
    cfg = get_config(args.model)
    with torch.device(device):
        target = LlamaForCausalLM(cfg).eval()
        draft = LlamaForCausalLM(
            get_config(args.model,
                       num_hidden_layers=args.draft_layers)).eval()
        heads = torch.nn.ModuleList(
            [MedusaHead(cfg.hidden_size, cfg.vocab_size)
             for _ in range(args.spec_len)])
    torch.set_default_dtype(prev)

    torch.manual_seed(1)
    x = torch.randint(0, cfg.vocab_size, (args.batch, args.prompt_len),
                      device=device)

    t0 = time.perf_counter()
    ref = generate(target, x, max_new_tokens=args.new_tokens)
    t_ref = time.perf_counter() - t0

    t0 = time.perf_counter()
    out, rate = speculative_generate(target, draft, x,
                                     max_new_tokens=args.new_tokens,
                                     spec_len=args.spec_len)
    t_spec = time.perf_counter() - t0
    assert torch.equal(out, ref), "speculative output diverged"

    t0 = time.perf_counter()
    out_m, rate_m = medusa_generate(target, heads, x,
                                    max_new_tokens=args.new_tokens)
    t_med = time.perf_counter() - t0
    assert torch.equal(out_m, ref), "medusa output diverged"

    print(f"target-only: {t_ref:.3f}s | draft-verify: {t_spec:.3f}s "
          f"(acceptance {rate:.2f}) | medusa: {t_med:.3f}s "
          f"(acceptance {rate_m:.2f}) — outputs identical")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
