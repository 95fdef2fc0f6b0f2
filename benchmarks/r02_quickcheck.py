"""Quick GPU correctness check for round-2 kernels (cls_pos, embed_pos,
layernorm_bwd H>1536, sigmoid determinism, 8p GEMM edge shapes)."""

import os, sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

import jimm_amd  # noqa: F401
from jimm_amd import ops
from jimm_amd.ops import _backend

dev = torch.device("cuda:0")
ext = _backend.ext()
ok = True


def check(name, a, b, tol):
    global ok
    err = (a.float() - b.float()).abs().max().item()
    good = err <= tol
    ok &= good
    print(f"{name:34s} maxerr {err:.3e}  {'OK' if good else 'FAIL'}", flush=True)


# K2 cls+pos
x = torch.randn(4, 196, 768, device=dev).bfloat16().requires_grad_(True)
cls = torch.randn(1, 1, 768, device=dev).bfloat16().requires_grad_(True)
pos = torch.randn(1, 197, 768, device=dev).bfloat16().requires_grad_(True)
y = ops.add_cls_pos(x, cls, pos)
ref = torch.cat([cls.expand(4, -1, -1), x], 1) + pos
check("cls_pos fwd", y, ref, 0)
g = torch.randn_like(y)
y.backward(g)
gx, gc, gp = x.grad.clone(), cls.grad.clone(), pos.grad.clone()
x.grad = cls.grad = pos.grad = None
ref2 = torch.cat([cls.expand(4, -1, -1), x], 1) + pos
ref2.backward(g)
check("cls_pos bwd dx", gx, x.grad, 0)
check("cls_pos bwd dcls", gc, cls.grad, 0)
check("cls_pos bwd dpos", gp, pos.grad, 0)

# no-cls variant
x2 = torch.randn(4, 196, 768, device=dev).bfloat16()
y2 = ops.add_cls_pos(x2, None, pos[:, :196].contiguous())
check("cls_pos fwd (no cls)", y2, x2 + pos[:, :196], 0)

# K10 embed+pos
ids = torch.randint(0, 1000, (4, 77), device=dev)
emb = torch.randn(1000, 512, device=dev).bfloat16().requires_grad_(True)
pe = torch.randn(1, 77, 512, device=dev).bfloat16().requires_grad_(True)
y = ops.embed_pos(ids, emb, pe)
ref = torch.nn.functional.embedding(ids, emb) + pe
check("embed_pos fwd", y, ref, 0)
g = torch.randn_like(y)
y.backward(g)
ge, gp = emb.grad.clone(), pe.grad.clone()
emb.grad = pe.grad = None
(torch.nn.functional.embedding(ids, emb) + pe).backward(g)
check("embed_pos bwd demb", ge, emb.grad, 6e-2)  # bf16 index_add order noise
check("embed_pos bwd dpos", gp, pe.grad, 0)

# LN bwd H=1664 (streaming path)
h = 1664
xl = torch.randn(512, h, device=dev, dtype=torch.float32, requires_grad=True)
wl = torch.randn(h, device=dev)
bl = torch.randn(h, device=dev)
yl = ops.layer_norm(xl, wl, bl, 1e-6)
gl = torch.randn_like(yl)
yl.backward(gl)
refl = torch.nn.functional.layer_norm(xl.detach().clone().requires_grad_(True), (h,), wl, bl, 1e-6)
xref = refl.grad_fn
x3 = xl.detach().clone().requires_grad_(True)
r3 = torch.nn.functional.layer_norm(x3, (h,), wl, bl, 1e-6)
r3.backward(gl)
check("ln_bwd H=1664 dx", xl.grad, x3.grad, 2e-4)

# sigmoid loss determinism
lg = torch.randn(256, 256, device=dev, dtype=torch.float32)
l1, _ = ext.sigmoid_loss_ew(lg, 0)
l2, _ = ext.sigmoid_loss_ew(lg, 0)
print(f"sigmoid loss deterministic: {bool((l1 == l2).all().item())}", flush=True)

# 8p GEMM ragged-M + acts
for M, N, K, act in [(777, 256, 128, ""), (4096, 768, 768, "gelu"), (1000, 512, 256, "quickgelu")]:
    x = (torch.rand(M, K, device=dev) * 2 - 1).bfloat16()
    w = ((torch.rand(N, K, device=dev) * 2 - 1) / K**0.5).bfloat16()
    b = torch.randn(N, device=dev).bfloat16()
    res = torch.randn(M, N, device=dev).bfloat16()
    y, z = ext.linear_fwd(x, w, b, act, res, True)
    zr = torch.nn.functional.linear(x.float(), w.float(), b.float())
    if act == "gelu":
        yr = torch.nn.functional.gelu(zr) + res.float()
    elif act == "quickgelu":
        yr = zr * torch.sigmoid(1.702 * zr) + res.float()
    else:
        yr = zr + res.float()
    scale = zr.abs().max().item()
    check(f"8p gemm M={M} N={N} K={K} {act or '-'} y", y, yr, 0.02 * scale)
    check(f"8p gemm M={M} N={N} K={K} {act or '-'} z", z, zr, 0.02 * scale)

print("ALL OK" if ok else "FAILURES", flush=True)
