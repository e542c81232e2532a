import sys
import torch
sys.path.insert(0, '.')
from sat_amd import _C
from sat_amd.ops import functional as F, hip

DEV = 'cuda'
torch.manual_seed(3)
B, L, A, D = 32, 196, 512, 512
t1 = torch.randn(B * L, A).to(DEV, torch.bfloat16)
t2 = torch.randn(B, A).to(DEV, torch.bfloat16)
v = (torch.randn(A) * 0.05).to(DEV, torch.bfloat16)
ctx = torch.randn(B, L, D).to(DEV, torch.bfloat16)
seed = torch.tensor(12345, dtype=torch.int64, device=DEV)


def rel(a, b):
    return ((a.float() - b.float()).abs().max() /
            b.float().abs().max().clamp_min(1e-6)).item()


tdrop, logits = _C.attn_scores_fused(t1, t2, v, seed, 0.0, 7, L)
tref = t1.float() + t2.float().repeat_interleave(L, dim=0)
lref = tref.matmul(v.float()).reshape(B, L)
print('tdrop rel:', rel(tdrop, tref))
print('logits rel:', rel(logits, lref))

alpha, pooled = _C.attn_pool_fwd(ctx, logits)
aref = torch.softmax(lref, dim=1)
pref = (ctx.float() * aref.unsqueeze(2)).sum(1)
print('alpha rel:', rel(alpha, aref))
print('pooled rel:', rel(pooled, pref))

# backward pieces
dalpha = torch.randn(B, L, device=DEV)
dpooled = torch.randn(B, D, device=DEV, dtype=torch.bfloat16)
dlogits, dctx = _C.attn_pool_bwd(ctx, alpha, dalpha, dpooled, True)
s = torch.bmm(ctx.float(), dpooled.float().unsqueeze(2)).squeeze(2)
da = dalpha + s
dlref = aref * (da - (aref * da).sum(1, keepdim=True))
print('dlogits rel:', rel(dlogits, dlref))
print('dctx rel:', rel(dctx, aref.unsqueeze(2) * dpooled.float().unsqueeze(1)))

dt1, dt2f, dvf = _C.attn_scores_bwd(tdrop, v, dlogits, seed, 0.0, 7, L)
dtref = (dlref.reshape(-1, 1) * v.float().unsqueeze(0))
print('dt1 rel:', rel(dt1, dtref))
print('dt2 rel:', rel(dt2f, dtref.reshape(B, L, A).sum(1)))
print('dv rel:', rel(dvf, tref.t().matmul(dlref.reshape(-1))))

# full autograd path
t1g = t1.clone().requires_grad_(True)
t2g = t2.clone().requires_grad_(True)
vg = v.clone().requires_grad_(True)
ctxg = ctx.clone().requires_grad_(True)
al, po = hip.attention_tail(t1g, t2g, vg, ctxg, 0.0, seed, 7)
refs = [t.detach().float().requires_grad_(True) for t in (t1g, t2g, vg, ctxg)]
ra, rp = F.attention_tail(*refs, 0.0, False)
print('full alpha rel:', rel(al, ra), 'pooled rel:', rel(po, rp))
ga = torch.randn_like(al)
gp = torch.randn_like(po)
(al * ga).sum().backward(retain_graph=True)
(po.float() * gp.float()).sum().backward()
(ra * ga).sum().backward(retain_graph=True)
(rp * gp.float()).sum().backward()
for name, t, r in zip('t1 t2 v ctx'.split(), (t1g, t2g, vg, ctxg), refs):
    print('grad', name, 'rel:', rel(t.grad, r.grad))

# dropout mask consistency
p = 0.5
tdA, lgA = _C.attn_scores_fused(t1, t2, v, seed, p, 3, L)
dlr = torch.randn(B, L, device=DEV)
dt1b, _, _ = _C.attn_scores_bwd(tdA, v, dlr, seed, p, 3, L)
nzf = (tdA != 0)
nzb = (dt1b != 0)
mismatch = (nzf != nzb)
# backward zero can also come from dlogit==0 or v==0; count fwd-dropped but bwd-nonzero
bad = (~nzf & nzb).sum().item()
print('fwd-dropped but bwd-nonzero:', bad, 'of', nzf.numel())
