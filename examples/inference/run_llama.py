#!/usr/bin/env python3
"""Llama inference on MI355X: shard a checkpoint, build the bucketed
NxDModel (hipGraph-captured decode), and generate.

The MI355X analogue of the reference's examples/inference/run_llama.py:
  torchrun --nproc-per-node 8 examples/inference/run_llama.py \
      --model llama3-8b --tp 8 --batch 32 --prompt-len 512 --new-tokens 128
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))

import torch
import torch.distributed as dist

import neuronx_distributed_amd as nxd
from neuronx_distributed_amd.models import get_config, LlamaForCausalLM
from neuronx_distributed_amd.inference import generate
from neuronx_distributed_amd.utils.sampling import Sampler


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama3-8b")
    p.add_argument("--tp", type=int, default=8)
    p.add_argument("--batch", type=int, default=32)
    p.add_argument("--prompt-len", type=int, default=512)
    p.add_argument("--new-tokens", type=int, default=128)
    p.add_argument("--top-k", type=int, default=0, help="0 = greedy")
    args = p.parse_args()

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29793")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
        dist.init_process_group("nccl")
    else:
        dist.init_process_group("gloo")
    nxd.parallel.initialize_model_parallel(tensor_model_parallel_size=args.tp)
    nxd.parallel.model_parallel_manual_seed(0)

    os.environ.setdefault("NXDA_FAST_INIT", "1")  # random weights demo
    cfg = get_config(args.model)
    dtype = torch.bfloat16 if torch.cuda.is_available() else torch.float32
    prev = torch.get_default_dtype()
    torch.set_default_dtype(dtype)
    model = LlamaForCausalLM(cfg).eval()
    torch.set_default_dtype(prev)
    if torch.cuda.is_available():
        model = model.cuda()

    device = next(model.parameters()).device
    torch.manual_seed(0)
    prompt = torch.randint(0, cfg.vocab_size, (args.batch, args.prompt_len),
                           device=device)
    sampler = Sampler(do_sample=args.top_k > 0, top_k=args.top_k or 50)

    with torch.no_grad():
        t0 = time.time()
        out = generate(model, prompt, max_new_tokens=args.new_tokens,
                       sampler=sampler)
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        dt = time.time() - t0
    if rank == 0:
        new = out.shape[1] - args.prompt_len
        print(f"generated {new} tokens x batch {args.batch} in {dt:.2f}s = "
              f"{args.batch * new / dt:,.0f} tokens/s")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
