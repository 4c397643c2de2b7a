"""Microbench: blockwise MoE forward paths on Mixtral-8x7B-class shapes.
Run on GPU: python tests/bench_blockwise.py"""
import time
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def bench(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000


def main():
    import torch.distributed as dist
    import os
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29771")
    if not dist.is_initialized():
        dist.init_process_group("nccl", rank=0, world_size=1)
    from neuronx_distributed_amd.parallel import parallel_state as ps
    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    from neuronx_distributed_amd.moe.blockwise import (blockwise_mm,
                                                       compute_block_indices)

    T, H, I, E, k = 8192, 4096, 14336, 8, 2
    dev = "cuda"
    torch.manual_seed(0)
    x = torch.randn(T, H, device=dev, dtype=torch.bfloat16)
    gu_w = torch.randn(E, H, 2 * I, device=dev, dtype=torch.bfloat16) * 0.02
    d_w = torch.randn(E, I, H, device=dev, dtype=torch.bfloat16) * 0.02
    logits = torch.randn(T, E, device=dev)
    vals, idx = torch.topk(torch.softmax(logits, -1), k, -1)
    aff = torch.zeros(T, E, device=dev).scatter(-1, idx, vals)

    BS = 512
    tpi, b2e, nb = compute_block_indices(idx, E, BS)
    print(f"blocks={nb}")

    t_cur = bench(lambda: blockwise_mm(x, aff, gu_w, d_w, tpi, b2e, idx, BS))
    flops = 2 * T * k * H * 2 * I + 2 * T * k * I * H  # gate_up + down
    print(f"current blockwise_mm: {t_cur:.3f} ms  ({flops/t_cur/1e9:.0f} TF/s)")

    # pure-GEMM floor: dense bmm of the same padded token count
    NB = b2e.numel()
    xb = torch.randn(NB * BS, H, device=dev, dtype=torch.bfloat16)
    def floor():
        gu = xb @ gu_w[0]
        act = torch.nn.functional.silu(gu[:, :I]) * gu[:, I:]
        return act @ d_w[0]
    t_floor = bench(floor)
    print(f"monolithic-GEMM floor (same padded tokens): {t_floor:.3f} ms")

    # grouped_mm path probe
    try:
        xg = xb.reshape(NB, BS, H)
        counts = torch.bincount(b2e, minlength=E)
        offs = torch.cumsum(counts * BS, 0, dtype=torch.int32)
        out = torch._grouped_mm(xb, gu_w, offs=offs)
        def grouped():
            gu = torch._grouped_mm(xb, gu_w, offs=offs)
            act = torch.nn.functional.silu(gu[:, :I]) * gu[:, I:]
            return torch._grouped_mm(act, d_w, offs=offs)
        t_g = bench(grouped)
        print(f"_grouped_mm path: {t_g:.3f} ms")
    except Exception as e:
        print("grouped_mm failed:", type(e).__name__, str(e)[:200])


if __name__ == "__main__":
    main()
