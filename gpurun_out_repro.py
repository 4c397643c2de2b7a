import sys, os
sys.path.insert(0, "/root/repo")
import torch
from neuronx_distributed_amd import ops
from neuronx_distributed_amd.ops import build
build.build()
torch.manual_seed(0)
for (B, Hq, Hkv, S) in [(8, 2, 2, 256), (8, 2, 2, 128), (4, 2, 2, 256),
                        (8, 4, 4, 256), (2, 2, 2, 256)]:
    q = torch.randn(B, Hq, S, 128, dtype=torch.bfloat16, device="cuda") * 0.5
    k = torch.randn(B, Hkv, S, 128, dtype=torch.bfloat16, device="cuda") * 0.5
    v = torch.randn(B, Hkv, S, 128, dtype=torch.bfloat16, device="cuda") * 0.5
    out = ops.flash_attn(q, k, v, causal=True)
    torch.cuda.synchronize()
    from neuronx_distributed_amd.kernels.flash_attn import _torch_reference
    ref = _torch_reference(q.float(), k.float(), v.float(), causal=True)
    err = (out.float() - ref).abs().max().item()
    print(f"B{B} Hq{Hq} S{S}: err {err:.4f}", flush=True)
print("ALL_OK")
