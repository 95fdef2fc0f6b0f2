"""Compare current kernel microbenches against the recorded round-1 floors.

Run on an MI355X:  python benchmarks/check_regressions.py
Exits nonzero if any measurement falls >10% below its recorded floor
(cross-box DVFS spread is ~4%; 10% indicates a real regression).
"""
import math
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

import jimm_amd  # noqa: F401
from jimm_amd.ops import _backend
from jimm_amd.ops._backend import maybe_enable_tunableop

# round-2 recorded values (TF/s), with a ~10% regression margin applied
# (r02: block-image attention + small-L kernel + in-house GEMM default)
FLOORS_ATTN = {  # (B, H, L, causal): (fwd_tf, bwd_tf)
    (256, 12, 197, False): (150, 155),
    (64, 16, 577, False): (265, 235),
    (256, 8, 77, True): (72, 62),     # strip-per-wave small-L fwd + fused small bwd
    (256, 12, 257, False): (195, 165),
}
FLOOR_STEP_VIT_B1024 = 5500  # img/s (r02 final: 5753)
# in-house GEMM floors on the b1024 model shapes (median TF/s, -10%)
FLOORS_GEMM = {  # (M, N, K, act): fwd_tf
    (201728, 2304, 768, ""): 790,
    (201728, 3072, 768, "gelu"): 600,
    (201728, 768, 3072, ""): 890,
}
FLOORS_DW = {  # (M, N, K): tf (post splitm rework + workspace combine)
    (201728, 2304, 768): 640,
    (201728, 3072, 768): 730,
    (201728, 768, 3072): 730,
}


def bench(fn, iters=20, warmup=6):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    maybe_enable_tunableop()
    ext = _backend.ext()
    dev = torch.device("cuda:0")
    failures = []
    for (B, H, L, causal), (f_fwd, f_bwd) in FLOORS_ATTN.items():
        q = torch.randn(B, H, L, 64, device=dev).bfloat16()
        k, v, do = torch.randn_like(q), torch.randn_like(q), torch.randn_like(q)
        scale = 1 / math.sqrt(64)
        o, lse = ext.attn_fwd(q, k, v, causal, scale)
        oc = o.contiguous()
        dq, dk, dvv = torch.empty_like(q), torch.empty_like(k), torch.empty_like(v)
        tf = 4 * B * H * L * L * 64 / 1e12 * (0.5 if causal else 1.0)
        r_fwd = tf / bench(lambda: ext.attn_fwd(q, k, v, causal, scale))
        r_bwd = 2.5 * tf / bench(lambda: ext.attn_bwd_fused(q, k, v, oc, do, lse, dq, dk, dvv, causal, scale))
        status_f = "OK" if r_fwd > 0.9 * f_fwd else "REGRESSION"
        status_b = "OK" if r_bwd > 0.9 * f_bwd else "REGRESSION"
        print(f"attn L={L:4d} causal={int(causal)}: fwd {r_fwd:6.1f} TF/s [{status_f}]  "
              f"bwd {r_bwd:6.1f} TF/s [{status_b}]", flush=True)
        if "REGRESSION" in (status_f, status_b):
            failures.append((B, H, L, causal))
    for (M, N, K, act), floor in FLOORS_GEMM.items():
        x = (torch.rand(M, K, device=dev) * 2 - 1).bfloat16()
        w = ((torch.rand(N, K, device=dev) * 2 - 1) / K**0.5).bfloat16()
        b = torch.randn(N, device=dev).bfloat16()
        tf = 2 * M * N * K / 1e12
        r = tf / bench(lambda: ext.linear_fwd(x, w, b, act, None, bool(act)), iters=10)
        st = "OK" if r > 0.9 * floor else "REGRESSION"
        print(f"gemm fwd M={M} N={N:5d} K={K:5d} {act or '-':5s}: {r:6.1f} TF/s [{st}]", flush=True)
        if st == "REGRESSION":
            failures.append((M, N, K, act))
    for (M, N, K), floor in FLOORS_DW.items():
        dz = (torch.rand(M, N, device=dev) * 2 - 1).bfloat16()
        x = ((torch.rand(M, K, device=dev) * 2 - 1) / 8).bfloat16()
        tf = 2 * M * N * K / 1e12
        r = tf / bench(lambda: ext.gemm_tn_8p(dz, x), iters=10)
        st = "OK" if r > 0.9 * floor else "REGRESSION"
        print(f"gemm dW  M={M} N={N:5d} K={K:5d}: {r:6.1f} TF/s [{st}]", flush=True)
        if st == "REGRESSION":
            failures.append((M, N, K))
    if failures:
        print("REGRESSIONS:", failures)
        sys.exit(1)
    print("ALL KERNEL FLOORS HELD")


if __name__ == "__main__":
    main()
