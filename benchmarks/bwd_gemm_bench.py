"""Microbench of the training-step GEMM family (fwd NT, dX NN, dW TN) via
rocBLAS/hipBLASLt, optionally under PyTorch TunableOp (hipBLASLt algo search,
including split-K — the round-1 profile showed Tensile picking a 27-workgroup
kernel for the dW shapes, ~0.4 PF/s on a 256-CU chip).

Usage (on an MI355X via gpurun):
    python benchmarks/bwd_gemm_bench.py             # baseline
    python benchmarks/bwd_gemm_bench.py --tune out.csv   # tune + save results
    python benchmarks/bwd_gemm_bench.py --load out.csv   # use saved tunings
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

# (M, N, K) of the forward x(M,K) @ w(N,K)^T — full model-zoo shape set
# (ViT-B/16@224 bs256, ViT-L/16@384 bs64, CLIP-B/32 bs256 both towers,
# SigLIP-base/16-256 bs256 both towers, patch embeds, projections)
SHAPES = [
    # ViT-B/16 @ 224, bs 256 (M = 256*197)
    (50432, 2304, 768), (50432, 768, 768), (50432, 3072, 768), (50432, 768, 3072),
    (50176, 768, 768),             # patch embed (M = 256*196)
    (256, 1000, 768),              # classifier
    # ViT-L/16 @ 384, bs 64 (M = 64*577)
    (36928, 3072, 1024), (36928, 1024, 1024), (36928, 4096, 1024), (36928, 1024, 4096),
    (36864, 1024, 768),            # patch embed
    # CLIP-B/32 vision bs 256 (M = 256*50) + text (M = 256*77)
    (12800, 2304, 768), (12800, 768, 768), (12800, 3072, 768), (12800, 768, 3072),
    (12544, 768, 3072),            # patch embed (K = 3*32*32)
    (19712, 1536, 512), (19712, 512, 512), (19712, 2048, 512), (19712, 512, 2048),
    (256, 512, 768), (256, 512, 512),   # visual/text projections
    # SigLIP-base/16-256 vision bs 256 (M = 256*256) + text (M = 256*64)
    (65536, 2304, 768), (65536, 768, 768), (65536, 3072, 768), (65536, 768, 3072),
    (16384, 2304, 768), (16384, 768, 768), (16384, 3072, 768), (16384, 768, 3072),
    (256, 768, 768),               # text head / MAP-head linears
    # default-batch shapes (vit b1024, clip b1024, siglip b512, vitl384 b128)
    (201728, 2304, 768), (201728, 768, 768), (201728, 3072, 768), (201728, 768, 3072),
    (200704, 768, 768),
    (73856, 3072, 1024), (73856, 1024, 1024), (73856, 4096, 1024), (73856, 1024, 4096),
    (73728, 1024, 768),
    (51200, 2304, 768), (51200, 768, 768), (51200, 3072, 768), (51200, 768, 3072),
    (50176, 768, 3072),
    (78848, 1536, 512), (78848, 512, 512), (78848, 2048, 512), (78848, 512, 2048),
    (1024, 512, 768), (1024, 512, 512), (1024, 1000, 768),
    (131072, 2304, 768), (131072, 768, 768), (131072, 3072, 768), (131072, 768, 3072),
    (32768, 2304, 768), (32768, 768, 768), (32768, 3072, 768), (32768, 768, 3072),
    (512, 768, 768),
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
    p = argparse.ArgumentParser()
    p.add_argument("--tune", metavar="CSV", default=None)
    p.add_argument("--load", metavar="CSV", default=None)
    args = p.parse_args()

    if args.tune or args.load:
        tun = torch.cuda.tunable
        if args.tune:
            tun.set_filename(args.tune)  # auto-written at process exit
            tun.enable(True)
            tun.tuning_enable(True)
            tun.set_max_tuning_duration(int(os.environ.get("TUNE_MS", "100")))
            tun.set_max_tuning_iterations(int(os.environ.get("TUNE_ITERS", "100")))
        else:
            tun.enable(True)
            tun.tuning_enable(False)
            assert tun.read_file(args.load), f"failed to read {args.load}"

    dev = torch.device("cuda:0")
    skip = int(os.environ.get("TUNE_SKIP", "0"))
    only = int(os.environ.get("TUNE_COUNT", str(len(SHAPES))))
    for M, N, K in SHAPES[skip:skip + only]:
        x = torch.randn(M, K, device=dev).bfloat16()
        w = torch.randn(N, K, device=dev).bfloat16() / K**0.5
        dz = torch.randn(M, N, device=dev).bfloat16()
        tf = 2 * M * N * K / 1e12

        t_fwd = bench(lambda: torch.matmul(x, w.t()))      # NT fwd
        t_dx = bench(lambda: torch.matmul(dz, w))          # NN dX
        t_dw = bench(lambda: torch.matmul(dz.t(), x))      # TN dW
        print(
            f"M={M:6d} N={N:5d} K={K:5d}  "
            f"fwd {tf/t_fwd:7.1f} TF/s ({t_fwd*1e3:6.2f} ms)  "
            f"dX {tf/t_dx:7.1f} TF/s ({t_dx*1e3:6.2f} ms)  "
            f"dW {tf/t_dw:7.1f} TF/s ({t_dw*1e3:6.2f} ms)",
            flush=True,
        )

    if args.tune:
        print("tuning results will be written at exit to", args.tune)


if __name__ == "__main__":
    main()
