"""tr16 lane-mapping probe + TN dW kernel correctness + bench + gradact."""

import os, statistics, sys, time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

import jimm_amd  # noqa: F401
from jimm_amd.ops import _backend

ext = _backend.ext()
dev = torch.device("cuda:0")

# ---- tr16 probe: LDS holds shorts 0..255; dump per-lane results ----------
src = torch.arange(256, dtype=torch.int16, device=dev)
for mode in (0, 1, 2):
    out = ext.tr16_probe(src, mode).cpu().numpy()
    print(f"tr16 mode {mode}: lane0 {out[0].tolist()} lane1 {out[1].tolist()} "
          f"lane15 {out[15].tolist()} lane16 {out[16].tolist()} lane17 {out[17].tolist()} "
          f"lane32 {out[32].tolist()}", flush=True)
# expected for the kernel's assumption (mode 0):
#   lane l gets column (l&15) of the [4][16] block at (l>>4)*128B:
#   lane l: [ (l>>4)*64 + (l&15), +16, +32, +48 ]

# ---- dW TN kernel correctness --------------------------------------------
torch.manual_seed(0)
ok = True
for M, N, K in [(4096, 256, 256), (5000, 512, 256), (50432, 2304, 768), (201728, 768, 3072)]:
    dz = (torch.rand(M, N, device=dev) * 2 - 1).bfloat16()
    x = ((torch.rand(M, K, device=dev) * 2 - 1) / 8).bfloat16()
    dw = ext.gemm_tn_8p(dz, x)
    ref = torch.matmul(dz.t().float(), x.float())
    err = (dw - ref).abs().max().item()
    rel = err / ref.abs().max().item()
    good = rel < 3e-2
    ok &= good
    print(f"dW M={M} N={N} K={K}: maxabs {err:.3e} rel {rel:.3e} {'OK' if good else 'FAIL'}", flush=True)

# ---- gradact correctness --------------------------------------------------
for act in ("gelu", "quickgelu"):
    M, N, K = 4096, 768, 3072
    dy = (torch.rand(M, K, device=dev) * 2 - 1).bfloat16()
    w = ((torch.rand(K, N, device=dev) * 2 - 1) / 32).bfloat16()  # original [out_f=K, in_f=N]
    wt = w.t().contiguous()  # (N, K)
    z = (torch.rand(M, N, device=dev) * 4 - 2).bfloat16()
    dz = ext.gemm_nt_8p_gradact(dy, wt, z, act)
    g = torch.matmul(dy.float(), w.float())
    zf = z.float()
    if act == "gelu":
        import math
        cdf = 0.5 * (1 + torch.erf(zf * 0.7071067811865476))
        pdf = 0.3989422804014327 * torch.exp(-0.5 * zf * zf)
        ref = g * (cdf + zf * pdf)
    else:
        s = torch.sigmoid(1.702 * zf)
        ref = g * (s + 1.702 * zf * s * (1 - s))
    rel = (dz.float() - ref).abs().max().item() / ref.abs().max().item()
    good = rel < 3e-2
    ok &= good
    print(f"gradact {act}: rel {rel:.3e} {'OK' if good else 'FAIL'}", flush=True)

# ---- dW bench vs rocBLAS (with TunableOp table already loaded) ------------
def bench(fn, iters=8):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    ts = []
    for _ in range(6):
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        ts.append((time.perf_counter() - t0) / iters)
    return statistics.median(ts)

for M, N, K in [(201728, 2304, 768), (201728, 768, 768), (201728, 3072, 768),
                (201728, 768, 3072), (78848, 1536, 512), (73856, 4096, 1024)]:
    dz = (torch.rand(M, N, device=dev) * 2 - 1).bfloat16()
    x = ((torch.rand(M, K, device=dev) * 2 - 1) / 8).bfloat16()
    tf = 2 * M * N * K / 1e12
    t_mine = bench(lambda: ext.gemm_tn_8p(dz, x))
    t_blas = bench(lambda: torch.matmul(dz.t(), x))
    print(f"dW M={M} N={N:5d} K={K:5d}: mine {tf/t_mine:7.1f} TF/s ({t_mine*1e3:6.2f}ms)  "
          f"rocBLAS {tf/t_blas:7.1f} TF/s ({t_blas*1e3:6.2f}ms)", flush=True)

print("ALL OK" if ok else "FAILURES", flush=True)
