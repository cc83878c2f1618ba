"""Print actual error magnitudes for each HIP op vs fp32 oracle (GPU box)."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from fengshen_amd.ops import functional as F
from fengshen_amd.ops import get_ext

def rel(a, b):
    return ((a.float()-b.float()).abs().max() / b.float().abs().max().clamp(min=1e-6)).item()

g = torch.Generator(device="cuda").manual_seed(0)
def rnd(*s, dt=torch.bfloat16):
    return torch.randn(*s, generator=g, device="cuda", dtype=torch.float32).to(dt)

# rms
x = rnd(4,33,1024); w = rnd(1024)
y, ir = get_ext().rms_norm_fwd(x, w, 1e-6)
ref = F.eager_rms_norm(x.float(), w.float(), 1e-6)
print("rms fwd rel", rel(y, ref), "invrms sample", ir[:3].tolist())
x32 = x.float(); var = x32.pow(2).mean(-1,keepdim=True)
print("invrms ref", torch.rsqrt(var+1e-6).flatten()[:3].tolist())

gy = rnd(4,33,1024)
gx, gw = get_ext().rms_norm_bwd(gy, x, w, ir)
x2 = x.float().requires_grad_(True); w2 = w.float().requires_grad_(True)
F.eager_rms_norm(x2, w2, 1e-6).backward(gy.float())
print("rms bwd gx rel", rel(gx, x2.grad), "gw rel", rel(gw, w2.grad))

# ln
x = rnd(6,17,768); w = rnd(768); b = rnd(768)
y, mu, istd = get_ext().layer_norm_fwd(x, w, b, 1e-5)
ref = torch.nn.functional.layer_norm(x.float(), (768,), w.float(), b.float(), 1e-5)
print("ln fwd rel", rel(y, ref))

# swiglu
x = rnd(64, 2816)
y = get_ext().swiglu_fwd(x)
gch, uch = x.float().chunk(2,-1)
ref = torch.nn.functional.silu(gch)*uch
print("swiglu fwd rel", rel(y, ref))
gy = rnd(64, 1408)
gp = get_ext().swiglu_bwd(gy, x)
x2 = x.float().requires_grad_(True)
g2,u2 = x2.chunk(2,-1)
(torch.nn.functional.silu(g2)*u2).backward(gy.float())
print("swiglu bwd rel", rel(gp, x2.grad))

# adamw
n = 1<<20
master = torch.randn(n, device="cuda"); refp = torch.nn.Parameter(master.clone())
grad = torch.randn(n, device="cuda", dtype=torch.bfloat16)
m = torch.zeros(n, device="cuda"); v = torch.zeros(n, device="cuda")
out = torch.empty(n, device="cuda", dtype=torch.bfloat16)
opt = torch.optim.AdamW([refp], lr=1e-3, betas=(0.9,0.999), eps=1e-8, weight_decay=0.01)
get_ext().fused_adamw(master, grad, m, v, out, 1e-3, 0.9, 0.999, 1e-8, 0.01, 1)
refp.grad = grad.float(); opt.step()
d = (master - refp.detach()).abs()
print("adamw maxdiff", d.max().item(), "at", d.argmax().item(), "n_bad>1e-5:", (d>1e-5).sum().item())
print("m sample", m[:3].tolist())
print("master[:3]", master[:3].tolist(), "ref[:3]", refp.detach()[:3].tolist())
print("tail check: master[-3:]", master[-3:].tolist(), "ref", refp.detach()[-3:].tolist())
