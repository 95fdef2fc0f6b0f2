"""Microbench: in-house MFMA GEMM (csrc/gemm.hip) vs rocBLAS (torch.matmul)
on the model-zoo hot shapes. Run on an MI355X via gpurun."""

import os, sys, time
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

import jimm_amd  # noqa: F401
from jimm_amd.ops import _backend

SHAPES = [
    # (M, N, K, act) — ViT-B/16 @ bs256 forward shapes
    (50432, 2304, 768, ""),        # QKV
    (50432, 768, 768, ""),         # proj
    (50432, 3072, 768, "gelu"),    # fc1
    (50432, 768, 3072, ""),        # fc2
    (50176, 768, 768, ""),         # patch embed GEMM
    (19712, 1536, 512, ""),        # CLIP text QKV (bs256, L77)
]


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ext = _backend.ext()
    dev = torch.device("cuda:0")
    for M, N, K, act in SHAPES:
        x = torch.randn(M, K, device=dev).bfloat16()
        w = torch.randn(N, K, device=dev).bfloat16() / K**0.5
        b = torch.randn(N, device=dev).bfloat16()
        tf = 2 * M * N * K / 1e12

        t_mine = bench(lambda: ext.linear_fwd(x, w, b, act, None, False))
        t_blas = bench(lambda: torch.nn.functional.linear(x, w, b))
        # correctness spot check
        y, _ = ext.linear_fwd(x, w, b, "", None, False)
        ref = torch.nn.functional.linear(x, w, b)
        err = (y.float() - ref.float()).abs().max().item() / ref.float().abs().max().item()
        print(
            f"M={M:6d} N={N:5d} K={K:5d} act={act or '-':9s} "
            f"mine {tf / t_mine:7.1f} TF/s ({t_mine * 1e3:6.2f} ms)  "
            f"rocBLAS {tf / t_blas:7.1f} TF/s ({t_blas * 1e3:6.2f} ms)  relerr {err:.2e}",
            flush=True,
        )


def main_dw():
    ext = _backend.ext()
    dev = torch.device("cuda:0")
    for M, N, K in [(50432, 2304, 768), (50432, 768, 768), (50432, 3072, 768), (50432, 768, 3072)]:
        dz = torch.randn(M, N, device=dev).bfloat16()
        x = torch.randn(M, K, device=dev).bfloat16() / K**0.5
        tf = 2 * M * N * K / 1e12
        t_mine = bench(lambda: ext.gemm_tn_8p(dz, x))
        t_blas = bench(lambda: torch.matmul(dz.t(), x))
        dw = ext.gemm_tn_8p(dz, x)
        ref = torch.matmul(dz.t().float(), x.float())
        err = (dw.float() - ref).abs().max().item() / ref.abs().max().item()
        print(f"dW M={M:6d} N={N:5d} K={K:5d} mine {tf/t_mine:7.1f} TF/s ({t_mine*1e3:6.2f} ms)  "
              f"rocBLAS {tf/t_blas:7.1f} TF/s ({t_blas*1e3:6.2f} ms)  relerr {err:.2e}", flush=True)


if __name__ == "__main__":
    main()
    main_dw()
