import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from neuronx_distributed_amd import ops
from neuronx_distributed_amd.ops import build as b
b.build()
from neuronx_distributed_amd.kernels.flash_attn import _torch_reference

for S in (32, 64, 128, 256):
    for causal in (False, True):
        torch.manual_seed(5)
        B, Hq, Hkv, D = 1, 1, 1, 128
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
        errs = {n: (a.float() - r).abs().max().item()
                for n, a, r in (("dq", q.grad, qr.grad), ("dk", k.grad, kr.grad),
                                ("dv", v.grad, vr.grad))}
        print(f"S={S} causal={causal}: " +
              " ".join(f"{n}={e:.4f}" for n, e in errs.items()))

# detailed dv map for the smallest failing case
S, causal = 64, False
torch.manual_seed(5)
q = (torch.randn(1, 1, S, 128, dtype=torch.bfloat16, device="cuda") * 0.5).requires_grad_(True)
k = (torch.randn(1, 1, S, 128, dtype=torch.bfloat16, device="cuda") * 0.5).requires_grad_(True)
v = (torch.randn(1, 1, S, 128, dtype=torch.bfloat16, device="cuda") * 0.5).requires_grad_(True)
out = ops.flash_attn(q, k, v, causal=causal)
dy = torch.randn_like(out)
out.backward(dy)
qr, kr, vr = (t.detach().float().requires_grad_(True) for t in (q, k, v))
ref = _torch_reference(qr, kr, vr, causal=causal)
ref.backward(dy.float())
err = (v.grad.float() - vr.grad)[0, 0]
per_row = err.abs().amax(dim=1)
print("dv per-row err:", [round(x, 3) for x in per_row.tolist()])
per_col = err.abs().amax(dim=0)
print("dv per-d err (every 8):", [round(per_col[i].item(), 3) for i in range(0, 128, 8)])
