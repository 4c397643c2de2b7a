import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from neuronx_distributed_amd.ops import _require_lib, _ptr, _stream
from neuronx_distributed_amd.ops import build as b

b.build()
lib = _require_lib()


def acc_row(r, hi):
    return (r & 3) + 8 * (r >> 2) + 4 * hi


torch.manual_seed(0)
P = torch.randn(32, 32, dtype=torch.float32).abs()  # [k][q]
dO = (torch.randn(32, 128) * 0.5).to(torch.bfloat16)

# pack P into accumulator layout: lane l holds col q=l&31, rows acc_row(r,hi)
pt_in = torch.zeros(64, 16, dtype=torch.float32)
for l in range(64):
    for r in range(16):
        pt_in[l, r] = P[acc_row(r, l >> 5), l & 31]

pt_g = pt_in.cuda()
do_g = dO.cuda()
dv_g = torch.zeros(32, 128, dtype=torch.float32, device="cuda")
lib.run_probe_dv(_ptr(pt_g), _ptr(do_g), _ptr(dv_g), _stream())
torch.cuda.synchronize()

# reference: dv = round_bf16(P) @ dO
ref = P.to(torch.bfloat16).float() @ dO.float()
err = (dv_g.cpu() - ref).abs()
print("probe_dv max err:", err.max().item(), "ref max:", ref.abs().max().item())
bad = (err > 0.05).nonzero()
print("n bad:", len(bad))
if len(bad):
    print("bad sample:", bad[:10].tolist())
    k0, d0 = bad[0].tolist()
    print("got", dv_g.cpu()[k0, d0].item(), "want", ref[k0, d0].item())
    # err structure
    print("per-k err:", [round(x, 3) for x in err.amax(1).tolist()])
    print("per-d8 err:", [round(err.amax(0)[i:i+8].max().item(), 3) for i in range(0, 128, 8)])
