import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time, math, torch
from fengshen_amd.ops import get_ext
ext = get_ext()
b, h, s, d = 8, 40, 2048, 128
g = torch.Generator(device="cuda").manual_seed(0)
q = torch.randn(b,h,s,d, generator=g, device="cuda").to(torch.bfloat16)
k = torch.randn(b,h,s,d, generator=g, device="cuda").to(torch.bfloat16)
v = torch.randn(b,h,s,d, generator=g, device="cuda").to(torch.bfloat16)
gy = torch.randn(b,h,s,d, generator=g, device="cuda").to(torch.bfloat16)
scale = 1/math.sqrt(d)
o, lse = ext.flash_attn_fwd(q, k, v, scale)
# direct ext bwd timing
for _ in range(2):
    ext.flash_attn_bwd(q, k, v, o, gy, lse, scale)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(10):
    dq, dk, dv = ext.flash_attn_bwd(q, k, v, o, gy, lse, scale)
torch.cuda.synchronize()
print(f"direct ext bwd: {(time.perf_counter()-t0)/10*1000:.2f} ms/iter")
# autograd path
q.requires_grad_(True); k.requires_grad_(True); v.requires_grad_(True)
from fengshen_amd.ops.flash import flash_attention
for _ in range(2):
    out = flash_attention(q, k, v, scale); torch.autograd.backward(out, gy)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(10):
    out = flash_attention(q, k, v, scale)
    torch.autograd.backward(out, gy)
torch.cuda.synchronize()
print(f"autograd fwd+bwd: {(time.perf_counter()-t0)/10*1000:.2f} ms/iter")
# phase-timed
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(10):
    out = flash_attention(q, k, v, scale)
torch.cuda.synchronize()
print(f"autograd fwd only: {(time.perf_counter()-t0)/10*1000:.2f} ms/iter")
