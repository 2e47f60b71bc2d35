import sys
sys.path.insert(0, "/root/repo")
import torch
from torchx_amd import ops
hip = ops.hip_ops(True)
x = (torch.randn(128, 256, device="cuda", dtype=torch.bfloat16) * 3)
s = (x.abs().amax().float() / 448).clamp(min=1e-10)
o, ot, amax = hip.fp8_cast_transpose(x, s, False)
ref = (x.float() / s).to(torch.float8_e4m3fn)
print("e4m3 row err:", (o.float() - ref.float()).abs().max().item())
print("e4m3 tr  err:", (ot.float() - ref.t().contiguous().float()).abs().max().item())
print("amax kernel vs torch:", amax.item(), x.abs().amax().item())
o2, ot2, am2 = hip.fp8_cast_transpose(x, s, True)
ref2 = (x.float() / s).to(torch.float8_e5m2)
print("e5m2 row err:", (o2.float() - ref2.float()).abs().max().item())
print("e5m2 tr  err:", (ot2.float() - ref2.t().contiguous().float()).abs().max().item())
# scaled_mm path end to end vs bf16
M, K, N = 256, 2048, 1024
a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
sa = (a.abs().amax().float()/448).clamp(min=1e-10); sw_ = (w.abs().amax().float()/448).clamp(min=1e-10)
a8, a8t, _ = hip.fp8_cast_transpose(a, sa, False)
w8, w8t, _ = hip.fp8_cast_transpose(w, sw_, False)
y = torch._scaled_mm(a8, w8.t(), scale_a=sa, scale_b=sw_, out_dtype=torch.bfloat16)
ref = a @ w.t()
print("smm rel err:", ((y.float()-ref.float()).abs().max()/ref.abs().max()).item())
print("has nan:", torch.isnan(y.float()).any().item())
