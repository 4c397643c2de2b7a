import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, time
from neuronx_distributed_amd import ops
shapes = [(32,4096,4096),(32,12288,4096),(32,28672,4096),(32,4096,14336),(32,128256,4096)]
for M,N,K in shapes:
    x = torch.randn(M,K,device='cuda',dtype=torch.bfloat16)
    w = torch.randn(N,K,device='cuda',dtype=torch.bfloat16)
    for fn,name in [(lambda: ops.skinny_linear(x,w),'skinny'),(lambda: x@w.t(),'blaslt')]:
        for _ in range(5): fn()
        torch.cuda.synchronize()
        t0=torch.cuda.Event(True);t1=torch.cuda.Event(True);t0.record()
        for _ in range(30): fn()
        t1.record(); torch.cuda.synchronize()
        us = t0.elapsed_time(t1)/30*1000
        gbs = N*K*2/ (us*1e-6) / 1e12
        print(f"{name} {M}x{N}x{K}: {us:7.1f} us  {gbs:5.2f} TB/s")
