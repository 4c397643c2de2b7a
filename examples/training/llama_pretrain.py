#!/usr/bin/env python3
"""Llama pretraining on MI355X — TP + ZeRO-1 (+ optional SP / PP).

The MI355X analogue of the reference's
examples/training/llama/tp_zero1_llama_hf_pretrain.py: one process per
GPU over RCCL, synthetic or user data, sharded checkpointing with resume.

Launch (8 GPUs):
  torchrun --nproc-per-node 8 --master-addr 127.0.0.1 \
      examples/training/llama_pretrain.py --model llama2-7b --tp 8 \
      --seq 4096 --global-batch 32 --steps 100 --ckpt-dir ckpts
"""

import argparse
import os
import time

import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))

import torch
import torch.distributed as dist

import neuronx_distributed_amd as nxd
from neuronx_distributed_amd.models import get_config, LlamaForCausalLM
from neuronx_distributed_amd.models.llama import LlamaDecoderLayer


def _single_process_defaults():
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29792")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    os.environ.setdefault("LOCAL_RANK", "0")


def main():
    _single_process_defaults()
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama2-7b")
    p.add_argument("--tp", type=int, default=8)
    p.add_argument("--pp", type=int, default=1)
    p.add_argument("--seq", type=int, default=4096)
    p.add_argument("--global-batch", type=int, default=32)
    p.add_argument("--microbatch", type=int, default=4)
    p.add_argument("--steps", type=int, default=100)
    p.add_argument("--lr", type=float, default=1.5e-4)
    p.add_argument("--sequence-parallel", action="store_true")
    p.add_argument("--activation-checkpoint", action="store_true")
    p.add_argument("--ckpt-dir", default=None)
    p.add_argument("--ckpt-interval", type=int, default=50)
    args = p.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
        dist.init_process_group("nccl")
    else:
        dist.init_process_group("gloo")

    cfg = nxd.neuronx_distributed_config(
        tensor_parallel_size=args.tp,
        pipeline_parallel_size=args.pp,
        pipeline_config={"transformer_layer_cls": LlamaDecoderLayer,
                         "num_microbatches":
                             args.global_batch // args.microbatch,
                         "input_names": ["input_ids", "labels"]}
        if args.pp > 1 else None,
        sequence_parallel=args.sequence_parallel,
        activation_checkpoint_config="full"
        if args.activation_checkpoint else None,
        optimizer_config={"zero_one_enabled": True, "grad_clipping": True,
                          "max_grad_norm": 1.0},
    )
    nxd.parallel.model_parallel_manual_seed(1234)

    model_cfg = get_config(
        args.model, sequence_parallel_enabled=args.sequence_parallel,
        max_position_embeddings=max(args.seq, 4096))
    dtype = torch.bfloat16 if torch.cuda.is_available() else torch.float32

    def model_fn():
        prev = torch.get_default_dtype()
        torch.set_default_dtype(dtype)
        try:
            return LlamaForCausalLM(model_cfg)
        finally:
            torch.set_default_dtype(prev)

    model = nxd.initialize_parallel_model(cfg, model_fn)
    opt = nxd.initialize_parallel_optimizer(
        cfg, torch.optim.AdamW, model.parameters(), lr=args.lr,
        betas=(0.9, 0.95), weight_decay=0.1)

    start_step = 0
    if args.ckpt_dir and nxd.has_checkpoint(args.ckpt_dir):
        uc = nxd.load_checkpoint(args.ckpt_dir, tag="latest_if_exists",
                                 model=model, optimizer=opt)
        start_step = (uc or {}).get("step", 0)

    device = torch.device("cuda", local_rank) \
        if torch.cuda.is_available() else torch.device("cpu")
    dp = nxd.parallel.get_data_parallel_size()
    mb_per_rank = max(1, args.global_batch // dp // args.microbatch)
    torch.manual_seed(42 + nxd.parallel.get_data_parallel_rank())

    # results.json metrics + moving-average seq/s throughput (reference
    # examples/.../training_utils.py:343-369 Throughput +
    # tp_zero1_llama_hf_pretrain.py:470-515 TrainingMetrics)
    from neuronx_distributed_amd.utils.training_metrics import (
        Metric, Throughput, TrainingMetrics)

    metrics = TrainingMetrics("results.json") if rank == 0 else None
    thr = Throughput(batch_size=args.global_batch // dp, world_size=dp,
                     grad_accum_usteps=1, moving_avg_window_size=10)
    for step in range(start_step, args.steps):
        opt.zero_grad()
        if args.pp > 1:
            x = torch.randint(0, model_cfg.vocab_size,
                              (args.global_batch // dp, args.seq),
                              device=device)
            loss = model.run_train(input_ids=x, labels=x)
        else:
            for _ in range(mb_per_rank):
                x = torch.randint(0, model_cfg.vocab_size,
                                  (args.microbatch, args.seq), device=device)
                loss = model(x, labels=x)
                (loss / mb_per_rank).backward()
        opt.step()
        seq_per_s = thr.get_throughput()
        if rank == 0 and step % 10 == 0:
            print(f"step {step}: loss {loss.item():.4f} "
                  f"({seq_per_s * args.seq:,.0f} tokens/s)", flush=True)
        if args.ckpt_dir and (step + 1) % args.ckpt_interval == 0:
            nxd.save_checkpoint(args.ckpt_dir, tag=str(step + 1), model=model,
                                optimizer=opt, user_content={"step": step + 1},
                                num_kept=2, async_save=True)
    if metrics is not None:
        metrics.store_parameters({"Model": args.model, "Steps": args.steps,
                                  "TP": args.tp, "PP": args.pp,
                                  "SeqLen": args.seq,
                                  "GlobalBatch": args.global_batch})
        metrics.store_metrics([
            Metric("Throughput", round(seq_per_s, 2), "seq/s"),
            Metric("Final loss", round(float(loss.detach()), 4)),
        ])
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
