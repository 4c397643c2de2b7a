"""Per-shape GEMM microbench for the llama2-7b TP1 training step.

Measures every hipBLASLt GEMM the bench model issues (fwd, dgrad, wgrad)
at M = global tokens 16384, prints achieved TFLOP/s per shape so slow
shapes can be targeted (TunableOp re-tune / layout swap).

Usage (GPU box):  python tests/bench_gemm.py [--tune]
  --tune : enable TunableOp online tuning and write an updated CSV to
           gpurun_out/tunableop_tuned.csv
"""

import argparse
import os
import sys

ap = argparse.ArgumentParser()
ap.add_argument("--tune", action="store_true")
ap.add_argument("--iters", type=int, default=20)
args = ap.parse_args()

csv_out = "gpurun_out/tunableop_tuned.csv"
if args.tune:
    os.environ["PYTORCH_TUNABLEOP_ENABLED"] = "1"
    os.environ["PYTORCH_TUNABLEOP_TUNING"] = "1"
    os.environ["PYTORCH_TUNABLEOP_FILENAME"] = csv_out
    os.environ.setdefault("PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS", "120")
    os.environ.setdefault("PYTORCH_TUNABLEOP_MAX_TUNING_ITERATIONS", "300")

import torch  # noqa: E402

if not args.tune:
    # read-only use of the shipped table
    table = os.path.join(os.path.dirname(__file__), "..", "profiles",
                         "tunableop_gfx950.csv")
    if os.path.exists(table):
        os.environ["PYTORCH_TUNABLEOP_ENABLED"] = "1"
        os.environ["PYTORCH_TUNABLEOP_FILENAME"] = table
        torch.cuda.tunable.enable(True)
        torch.cuda.tunable.tuning_enable(False)
        torch.cuda.tunable.read_file(table)

M = 16384
H, I, V = 4096, 11008, 32000
shapes = [
    # (name, m, n, k)  out = (m,n), a (m,k) @ b (k,n)
    ("qkv_fused_fwd", M, 3 * H, H),
    ("q_fwd", M, H, H),
    ("o_fwd", M, H, H),
    ("gate_up_fwd", M, 2 * I, H),
    ("down_fwd", M, H, I),
    ("lm_head_fwd", M, V, H),
    # dgrad: dy (m,n) @ w (n,k) -> (m,k)
    ("qkv_dgrad", M, H, 3 * H),
    ("gate_up_dgrad", M, H, 2 * I),
    ("down_dgrad", M, I, H),
    ("lm_head_dgrad", M, H, V),
    # wgrad: dy^T (n,m) @ x (m,k) -> (n,k); a is a TRANSPOSED view
    ("qkv_wgrad", 3 * H, H, M),
    ("gate_up_wgrad", 2 * I, H, M),
    ("down_wgrad", H, I, M),
    ("lm_head_wgrad", V, H, M),
]

dev = "cuda"
results = []
for name, m, n, k in shapes:
    if "wgrad" in name:
        # dy (M tokens, m) and x (M tokens, k): grad = dy.t() @ x
        dy = torch.randn(k, m, dtype=torch.bfloat16, device=dev)
        x = torch.randn(k, n, dtype=torch.bfloat16, device=dev)
        fn = lambda: dy.t() @ x  # noqa: E731
    else:
        a = torch.randn(m, k, dtype=torch.bfloat16, device=dev)
        # fwd/dgrad in the layers: x @ w.t() with w (n,k) row-major
        w = torch.randn(n, k, dtype=torch.bfloat16, device=dev)
        fn = lambda: a @ w.t()  # noqa: E731
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(True)
    t1 = torch.cuda.Event(True)
    t0.record()
    for _ in range(args.iters):
        fn()
    t1.record()
    torch.cuda.synchronize()
    ms = t0.elapsed_time(t1) / args.iters
    tf = 2.0 * m * n * k / (ms * 1e-3) / 1e12
    results.append((name, m, n, k, ms, tf))
    print(f"{name:18s} m={m:6d} n={n:6d} k={k:6d}  {ms:7.3f} ms  {tf:7.0f} TF")

# wgrad alternative: materialize dy^T then NN GEMM (vs TN on a view)
for name, mm, nn, kk in [("gu_wgrad_xposNN", 2 * I, H, M),
                         ("lm_wgrad_xposNN", V, H, M),
                         ("dn_wgrad_xposNN", H, I, M)]:
    dy = torch.randn(kk, mm, dtype=torch.bfloat16, device=dev)
    x = torch.randn(kk, nn, dtype=torch.bfloat16, device=dev)
    fn = lambda: dy.t().contiguous() @ x  # noqa: E731
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(True); t1 = torch.cuda.Event(True)
    t0.record()
    for _ in range(args.iters):
        fn()
    t1.record(); torch.cuda.synchronize()
    ms = t0.elapsed_time(t1) / args.iters
    tf = 2.0 * mm * nn * kk / (ms * 1e-3) / 1e12
    print(f"{name:18s} m={mm:6d} n={nn:6d} k={kk:6d}  {ms:7.3f} ms  {tf:7.0f} TF (incl. transpose)")

tot = sum(2.0 * m * n * k for _, m, n, k, _, _ in results)
ttime = sum(ms for *_, ms, _ in results)
print(f"== aggregate {tot/ttime/1e9:7.0f} TF over {ttime:.1f} ms")
if args.tune:
    torch.cuda.tunable.write_file(csv_out)
    print("wrote", csv_out)
