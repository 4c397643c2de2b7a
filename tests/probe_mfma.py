"""Run the MFMA layout probes on GPU and print the derived maps."""
import ctypes
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from neuronx_distributed_amd.ops import _require_lib, _ptr, _stream
from neuronx_distributed_amd.ops import build as ops_build

ops_build.build()
lib = _require_lib()

# --- probe_c: D[m][n] = m*32 + n ------------------------------------------
A = torch.zeros(32, 16, dtype=torch.bfloat16, device="cuda")
B = torch.zeros(16, 32, dtype=torch.bfloat16, device="cuda")
A[:, 0] = torch.arange(32, dtype=torch.bfloat16)
A[:, 1] = 1.0
B[0, :] = 32.0
B[1, :] = torch.arange(32, dtype=torch.bfloat16)
out = torch.zeros(64, 16, dtype=torch.float32, device="cuda")
lib.run_probe_c(_ptr(A.contiguous()), _ptr(B.contiguous()), _ptr(out),
                _stream())
torch.cuda.synchronize()
o = out.cpu().numpy().astype(int)
# derive (row, col) per (lane, reg)
print("C-map: value = m*32+n; showing m,n for lanes 0,1,32,33 all regs:")
for l in (0, 1, 2, 32, 33):
    pairs = [(int(v) // 32, int(v) % 32) for v in o[l]]
    print(f" lane {l}: {pairs}")

# verify candidate formula
ok = True
for l in range(64):
    for r in range(16):
        m_pred = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5)
        n_pred = l & 31
        v = o[l][r]
        if v != m_pred * 32 + n_pred:
            ok = False
print("C-map matches (r&3)+8*(r>>2)+4*hi, col=l&31:", ok)

# --- probe_k: reveal j -> true-k relation ---------------------------------
# A elem J0=1 else 0; B elem j = 2^j.  D[m][n] = sum over {true k of A's
# elem J0 across its two owner lanes} of 2^{j_B(k,n)}.
for J0 in range(8):
    out.zero_()
    lib.run_probe_k(_ptr(out), ctypes.c_int(J0), _stream())
    torch.cuda.synchronize()
    o = out.cpu().numpy()
    # look at D[0][0] ... value is sum of powers of two = bitmask of B elems
    # whose true k matches A-elem-J0's true k (for each owner half)
    print(f"J0={J0}: D[0][0]={o[0][0]:.0f}  D[4][0]={o[2][0] if False else 0}"
          f"  lane0 regs={[int(x) for x in o[0][:8]]}")
