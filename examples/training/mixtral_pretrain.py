#!/usr/bin/env python3
"""Mixtral-style MoE pretraining example (mirrors llama_pretrain.py; the
reference's MoE training examples).  Launch:

  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
      --master-addr 127.0.0.1 examples/training/mixtral_pretrain.py \
      --model mixtral-8x7b --tp 8 --steps 100
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))

import torch
import torch.distributed as dist


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="tiny-moe")
    p.add_argument("--tp", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--batch", type=int, default=4)
    p.add_argument("--seq", type=int, default=256)
    p.add_argument("--lr", type=float, default=1.5e-4)
    args = p.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29781")
    on_gpu = torch.cuda.is_available()
    if on_gpu:
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)))
    dist.init_process_group("nccl" if on_gpu else "gloo", rank=rank,
                            world_size=world)

    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.models import (MixtralForCausalLM,
                                                get_moe_config)
    from neuronx_distributed_amd.optimizer import NeuronZero1Optimizer
    from neuronx_distributed_amd.utils.training_metrics import Throughput

    ps.initialize_model_parallel(tensor_model_parallel_size=args.tp)
    cfg = get_moe_config(args.model, max_position_embeddings=args.seq)
    device = torch.device("cuda") if on_gpu else torch.device("cpu")
    dtype = torch.bfloat16 if on_gpu else torch.float32
    prev = torch.get_default_dtype()
    torch.set_default_dtype(dtype)
    torch.manual_seed(0)
    with torch.device(device):
        model = MixtralForCausalLM(cfg)
    torch.set_default_dtype(prev)

    opt = NeuronZero1Optimizer(model.parameters(), torch.optim.AdamW,
                               lr=args.lr, grad_clipping=True, max_norm=1.0)
    tp_meter = Throughput(args.batch, world // args.tp)
    torch.manual_seed(1234 + rank)
    for step in range(args.steps):
        x = torch.randint(0, cfg.vocab_size, (args.batch, args.seq),
                          device=device)
        opt.zero_grad()
        loss = model(x, labels=x)
        loss.backward()
        opt.step()
        if rank == 0:
            print(f"step {step} loss {loss.item():.4f} "
                  f"{tp_meter.get_throughput():.1f} seq/s", flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
