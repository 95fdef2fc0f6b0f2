"""Within-probe interleaved A/B of the GEMM engines on the model-zoo shapes.

Variants: 8p (gemm8p.hip, 8-phase counted-vmcnt), 2ph (gemm256.hip,
vmcnt(0) 2-phase), blas (hipBLASLt via F.linear).  Interleaved rounds in one
process (guide §5.4 rule 24); reports median and min ms per variant.

Run: gpurun -- 'python benchmarks/gemm8p_ab.py'
"""

import os, statistics, sys, time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

import jimm_amd  # noqa: F401
from jimm_amd.ops import _backend

SHAPES = [
    # (M, N, K, act) — b1024 ViT-B/16 training shapes (the bench config)
    (201728, 2304, 768, ""),
    (201728, 768, 768, ""),
    (201728, 3072, 768, "gelu"),
    (201728, 768, 3072, ""),
    (200704, 768, 768, ""),  # patch embed GEMM b1024
    # b256 shapes (profile shapes)
    (50432, 2304, 768, ""),
    (50432, 3072, 768, "gelu"),
    (50432, 768, 3072, ""),
    # CLIP text b1024 (L77, w512)
    (78848, 1536, 512, ""),
    (78848, 2048, 512, "quickgelu"),
    (78848, 512, 2048, ""),
    # ViT-L/16-384 b128 (L577, w1024) — ragged M (577*128 = 73856 = 288.5*256)
    (73856, 3072, 1024, ""),
    (73856, 4096, 1024, "gelu"),
    (73856, 1024, 4096, ""),
    # square reference point (guide: 8-phase ~1330 TF/s @4k random)
    (4096, 4096, 4096, ""),
    (8192, 8192, 8192, ""),
]

ROUNDS = 8
ITERS = 6


def time_once(fn):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(ITERS):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / ITERS


def main():
    ext = _backend.ext()
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    for M, N, K, act in SHAPES:
        x = (torch.rand(M, K, device=dev, dtype=torch.float32) * 2 - 1).bfloat16()
        w = ((torch.rand(N, K, device=dev, dtype=torch.float32) * 2 - 1) / K**0.5).bfloat16()
        b = torch.randn(N, device=dev).bfloat16()
        tf = 2 * M * N * K / 1e12

        sup8 = ext.gemm8p_supported(M, N, K)
        sup2 = (M % 256 == 0) and (N % 256 == 0) and (K % 64 == 0)

        os.environ["JIMM_AMD_GEMM_TILE"] = "0"
        f8 = lambda: ext.linear_fwd(x, w, b, act, None, False)
        fb = lambda: torch.nn.functional.linear(x, w, b)

        def f2():
            os.environ["JIMM_AMD_GEMM_TILE"] = "256"
            try:
                return ext.linear_fwd(x, w, b, act, None, False)
            finally:
                os.environ["JIMM_AMD_GEMM_TILE"] = "0"

        # correctness first (8p vs fp32 reference)
        if sup8:
            y, _ = ext.linear_fwd(x, w, b, "", None, False)
            ref = torch.nn.functional.linear(x.float(), w.float(), b.float())
            err = (y.float() - ref).abs().max().item() / ref.abs().max().item()
        else:
            err = float("nan")

        res = {"8p": [], "2ph": [], "blas": []}
        for _ in range(2):  # warmup all variants
            if sup8:
                f8()
            if sup2:
                f2()
            fb()
        for _ in range(ROUNDS):
            if sup8:
                res["8p"].append(time_once(f8))
            if sup2:
                res["2ph"].append(time_once(f2))
            res["blas"].append(time_once(fb))

        def fmt(ts):
            if not ts:
                return "      --      "
            med = statistics.median(ts)
            return f"{tf / med:7.1f} TF/s ({med * 1e3:6.2f}ms)"

        print(
            f"M={M:6d} N={N:5d} K={K:5d} {act or '-':9s} "
            f"8p {fmt(res['8p'])}  2ph {fmt(res['2ph'])}  blas {fmt(res['blas'])}  "
            f"relerr {err:.2e}",
            flush=True,
        )


if __name__ == "__main__":
    main()
