import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time, math, torch
from fengshen_amd.ops.flash import flash_attention

b, h, s, d = 8, 40, 2048, 128
g = torch.Generator(device="cuda").manual_seed(0)
q = torch.randn(b,h,s,d, generator=g, device="cuda").to(torch.bfloat16)
k = torch.randn(b,h,s,d, generator=g, device="cuda").to(torch.bfloat16)
v = torch.randn(b,h,s,d, generator=g, device="cuda").to(torch.bfloat16)
scale = 1/math.sqrt(d)
# correctness spot-check at s=512 slice
qs, ks, vs = q[:1,:4,:512], k[:1,:4,:512], v[:1,:4,:512]
out = flash_attention(qs.contiguous(), ks.contiguous(), vs.contiguous(), scale)
sc = (qs.float() @ ks.float().transpose(-1,-2)) * scale
causal = torch.ones(512,512,device="cuda",dtype=torch.bool).triu(1)
ref = torch.softmax(sc.masked_fill(causal, float("-inf")), -1) @ vs.float()
err = (out.float()-ref).abs().max()/ref.abs().max()
print("refcheck rel err:", err.item())
for _ in range(3):
    flash_attention(q, k, v, scale)
torch.cuda.synchronize()
t0 = time.perf_counter(); N = 10
for _ in range(N):
    flash_attention(q, k, v, scale)
torch.cuda.synchronize()
dt = (time.perf_counter()-t0)/N
flops = 2*2*b*h*s*s*d*0.5
print(f"flash fwd: {dt*1000:.2f} ms, {flops/dt/1e12:.1f} TF")

# bwd timing
q.requires_grad_(True); k.requires_grad_(True); v.requires_grad_(True)
out = flash_attention(q, k, v, scale)
gy = torch.randn_like(out)
for _ in range(3):  # warm the allocator (cold hipMallocs dominate otherwise)
    out = flash_attention(q, k, v, scale)
    torch.autograd.backward(out, gy)
torch.cuda.synchronize()
t0 = time.perf_counter(); N2 = 5
for _ in range(N2):
    out = flash_attention(q, k, v, scale)
    torch.autograd.backward(out, gy)
torch.cuda.synchronize()
dt = (time.perf_counter()-t0)/N2
fb_flops = flops * 3.5  # fwd + bwd(2.5x)
print(f"flash fwd+bwd: {dt*1000:.2f} ms, {fb_flops/dt/1e12:.1f} TF-equiv")

# v3 fwd timing (same shape)
from fengshen_amd.ops import get_ext
qd, kd, vd = q.detach(), k.detach(), v.detach()
o3, lse3 = get_ext().flash_attn_fwd_v3(qd, kd, vd, scale, True, None, 0.0, 0)
o1 = flash_attention(qd, kd, vd, scale)
err3 = (o3.float() - o1.float()).abs().max() / o1.float().abs().max()
print("v3 vs v2 rel err:", err3.item())
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(N):
    get_ext().flash_attn_fwd_v3(qd, kd, vd, scale, True, None, 0.0, 0)
torch.cuda.synchronize()
dt3 = (time.perf_counter() - t0) / N
print(f"flash fwd v3: {dt3*1000:.2f} ms, {flops/dt3/1e12:.1f} TF")

# v3 fwd+bwd timing (routing picks v3 for causal d=128)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(N2):
    out = flash_attention(q, k, v, scale)
    torch.autograd.backward(out, gy)
torch.cuda.synchronize()
dt4 = (time.perf_counter() - t0) / N2
print(f"flash fwd+bwd (v3 routed): {dt4*1000:.2f} ms, {flops*3.5/dt4/1e12:.1f} TF-equiv")

# long-context scaling (the reference fused softmax capped at sk<=2048)
for s_long in (4096, 8192):
    bl = 2
    ql = torch.randn(bl, 40, s_long, 128, device="cuda").to(torch.bfloat16)
    kl = torch.randn(bl, 40, s_long, 128, device="cuda").to(torch.bfloat16)
    vl = torch.randn(bl, 40, s_long, 128, device="cuda").to(torch.bfloat16)
    for _ in range(2):
        flash_attention(ql, kl, vl, scale)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(5):
        flash_attention(ql, kl, vl, scale)
    torch.cuda.synchronize()
    dtl = (time.perf_counter() - t0) / 5
    fl = 2 * 2 * bl * 40 * s_long * s_long * 128 * 0.5
    print(f"flash fwd s={s_long}: {dtl*1000:.2f} ms, {fl/dtl/1e12:.1f} TF")
