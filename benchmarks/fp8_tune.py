"""TunableOp tuning for the fp8 scaled_mm signatures the fused block emits
(config 5, M=128*577): e4m3 x e4m3^T -> bf16, per-tensor scales, with and
without the fused bias epilogue. Writes gpurun_out/tunableop_fp8.csv whose
ScaledGemmTunableOp rows get merged into jimm_amd/data/tunableop_mi355x.csv.

Run on the GPU box:  python benchmarks/fp8_tune.py
"""

import os
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

M = int(os.environ.get("FP8_TUNE_M", 128 * 577))
SHAPES = [tuple(int(v) for v in p.split("x")) for p in os.environ.get(
    "FP8_TUNE_SHAPES", "3072x1024,4096x1024,1024x4096").split(",")]


def t_ms(fn, iters=30):
    s = torch.cuda.Event(True)
    e = torch.cuda.Event(True)
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters


def main():
    os.makedirs("gpurun_out", exist_ok=True)
    tun = torch.cuda.tunable
    tun.set_filename("gpurun_out/tunableop_fp8.csv")  # flushed at process exit
    tun.enable(True)
    tun.tuning_enable(True)
    for name, val in (("set_max_tuning_duration", 500), ("set_max_tuning_iterations", 200)):
        try:
            getattr(tun, name)(val)
        except AttributeError:
            pass

    cases = []
    for n, k in SHAPES:
        x8 = (torch.randn(M, k, device="cuda") / 4).to(torch.float8_e4m3fn)
        w8 = (torch.randn(n, k, device="cuda") / 30).to(torch.float8_e4m3fn)
        s = torch.ones(1, 1, device="cuda")
        b = torch.randn(n, device="cuda").bfloat16()
        for bias in (b, None):
            cases.append((n, k, x8, w8, s, bias))

    for n, k, x8, w8, s, bias in cases:
        torch._scaled_mm(x8, w8.t(), scale_a=s, scale_b=s, bias=bias,
                         out_dtype=torch.bfloat16)  # triggers tuning
        torch.cuda.synchronize()
        print(f"tuned ({n},{k}) bias={bias is not None}", flush=True)

    tun.tuning_enable(False)

    print(f"{'shape':16s} {'bias':5s} {'TF/s':>8s}")
    for n, k, x8, w8, s, bias in cases:
        ms = t_ms(lambda: torch._scaled_mm(x8, w8.t(), scale_a=s, scale_b=s,
                                           bias=bias, out_dtype=torch.bfloat16))
        print(f"({n},{k})".ljust(16) + f" {str(bias is not None):5s} "
              f"{2.0 * M * n * k / (ms * 1e-3) / 1e12:8.0f}")


if __name__ == "__main__":
    main()
