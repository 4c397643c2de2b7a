import torch
dev='cuda'
M,N,K=8192,4096,4096
a=torch.randn(M,K,device=dev).to(torch.float8_e4m3fn)
b=torch.randn(N,K,device=dev).to(torch.float8_e4m3fn)
sa=torch.tensor(1.0,device=dev); sb=torch.tensor(1.0,device=dev)
try:
    out=torch._scaled_mm(a,b.t(),scale_a=sa,scale_b=sb,out_dtype=torch.bfloat16)
    print("scaled_mm OK", out.shape, out.dtype)
    t0=torch.cuda.Event(True);t1=torch.cuda.Event(True)
    for _ in range(3): torch._scaled_mm(a,b.t(),scale_a=sa,scale_b=sb,out_dtype=torch.bfloat16)
    torch.cuda.synchronize(); t0.record()
    for _ in range(20): torch._scaled_mm(a,b.t(),scale_a=sa,scale_b=sb,out_dtype=torch.bfloat16)
    t1.record(); torch.cuda.synchronize()
    ms=t0.elapsed_time(t1)/20
    print(f"fp8 gemm {ms:.3f} ms = {2*M*N*K/ms/1e9:.0f} TF")
except Exception as e:
    print("scaled_mm FAIL:", e)
# rowwise scales
try:
    sa=torch.ones(M,1,device=dev); sb=torch.ones(1,N,device=dev)
    out=torch._scaled_mm(a,b.t(),scale_a=sa,scale_b=sb,out_dtype=torch.bfloat16)
    print("rowwise scaled_mm OK")
except Exception as e:
    print("rowwise FAIL:", repr(e)[:120])
# bf16 reference same shape
ab=torch.randn(M,K,device=dev,dtype=torch.bfloat16); bb=torch.randn(N,K,device=dev,dtype=torch.bfloat16)
for _ in range(3): ab@bb.t()
torch.cuda.synchronize()
t0=torch.cuda.Event(True);t1=torch.cuda.Event(True); t0.record()
for _ in range(20): ab@bb.t()
t1.record(); torch.cuda.synchronize()
ms=t0.elapsed_time(t1)/20
print(f"bf16 gemm {ms:.3f} ms = {2*M*N*K/ms/1e9:.0f} TF")
