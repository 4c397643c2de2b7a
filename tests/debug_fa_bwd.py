import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import math
import torch
from neuronx_distributed_amd import ops
from neuronx_distributed_amd.ops import build as b
b.build()
from neuronx_distributed_amd.kernels.flash_attn import _torch_reference

torch.manual_seed(5)
B, Hq, Hkv, S, D = 1, 2, 2, 256, 128
causal = True
q = (torch.randn(B, Hq, S, D, dtype=torch.bfloat16, device="cuda") * 0.5).requires_grad_(True)
k = (torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device="cuda") * 0.5).requires_grad_(True)
v = (torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device="cuda") * 0.5).requires_grad_(True)
out = ops.flash_attn(q, k, v, causal=causal)
dy = torch.randn_like(out)
out.backward(dy)

qr = q.detach().float().requires_grad_(True)
kr = k.detach().float().requires_grad_(True)
vr = v.detach().float().requires_grad_(True)
ref = _torch_reference(qr, kr, vr, causal=causal)
ref.backward(dy.float())

for name, a, r in (("dq", q.grad, qr.grad), ("dk", k.grad, kr.grad),
                   ("dv", v.grad, vr.grad), ("out", out, ref)):
    a = a.float()
    err = (a - r).abs()
    print(f"{name}: max_err={err.max().item():.4f} ref_absmax={r.abs().max().item():.4f} "
          f"mean_err={err.mean().item():.5f}")
    # per-32-row-block max err
    e = err.amax(dim=(0, 1, 3))
    blocks = [round(e[i:i+32].max().item(), 3) for i in range(0, S, 32)]
    print(f"   per-32row-block max: {blocks}")
    # per-d-block
    ed = err.amax(dim=(0, 1, 2))
    print(f"   per-32d-block max: {[round(ed[i:i+32].max().item(),3) for i in range(0,D,32)]}")
