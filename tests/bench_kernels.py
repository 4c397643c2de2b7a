import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from neuronx_distributed_amd import ops
from neuronx_distributed_amd.ops import build as b
b.build()

def timeit(fn, n=10, warm=3):
    for _ in range(warm): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1000

B, Hq, Hkv, S, D = 1, 32, 32, 4096, 128
q = torch.randn(B, Hq, S, D, dtype=torch.bfloat16, device="cuda") * 0.3
k = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device="cuda") * 0.3
v = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device="cuda") * 0.3

qg = q.clone().requires_grad_(True)
kg = k.clone().requires_grad_(True)
vg = v.clone().requires_grad_(True)
out = ops.flash_attn(qg, kg, vg, causal=True)
dy = torch.randn_like(out)

fwd_ms = timeit(lambda: ops.flash_attn(q, k, v, causal=True))
def bwd():
    o = ops.flash_attn(qg, kg, vg, causal=True)
    torch.autograd.backward(o, dy)
full_ms = timeit(bwd)
flops_fwd = 4 * S * S * D * Hq * B * 0.5
print(f"flash fwd: {fwd_ms:.2f} ms = {flops_fwd/fwd_ms*1e-9:.0f} TF")
print(f"fwd+bwd:  {full_ms:.2f} ms  (bwd ~{full_ms-fwd_ms:.2f} ms = "
      f"{2.5*flops_fwd/(full_ms-fwd_ms)*1e-9:.0f} TF)")

# rmsnorm
x = torch.randn(4096, 4096, dtype=torch.bfloat16, device="cuda")
w = torch.randn(4096, dtype=torch.bfloat16, device="cuda")
ms = timeit(lambda: ops.rmsnorm(x, w, 1e-5))
print(f"rmsnorm fwd 4096x4096: {ms:.3f} ms = {2*x.numel()*2/ms*1e-6:.0f} GB/s")
