"""Round-2 follow-up A/Bs at config-5 shapes (M=128*577):
  1. fc2-dX gradact: fused epilogue (bf16 / fp8-emit) vs plain GEMM +
     separate act_bwd (+ fp8_cast) passes.
  2. dW gemm_tn_8p across the 4 block shapes (run under different
     JIMM_AMD_DW_SPLITM to sweep the split factor — it is read once).
  3. layernorm_fwd vs layernorm_fwd_fp8 at (M, 1024).
Run on the GPU box: python benchmarks/r2_ab2.py [gradact|dw|ln|all]
"""

import os
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
from jimm_amd.ops import _backend  # noqa: E402

M = 128 * 577


def t_ms(fn, iters=20):
    s = torch.cuda.Event(True)
    e = torch.cuda.Event(True)
    for _ in range(4):
        fn()
    torch.cuda.synchronize()
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters


def gradact():
    ext = _backend.ext()
    dy = torch.randn(M, 1024, device="cuda").bfloat16()
    wt = torch.randn(4096, 1024, device="cuda").bfloat16() / 30
    z = torch.randn(M, 4096, device="cuda").bfloat16()
    s8 = torch.ones(1, device="cuda")
    am = torch.zeros(1, device="cuda")
    tf = 2.0 * M * 4096 * 1024 / 1e12

    def fused():
        return ext.gemm_nt_8p_gradact(dy, wt, z, "gelu")

    def fused8():
        return ext.gemm_nt_8p_gradact_fp8(dy, wt, z, "gelu", s8, am)

    def plain():
        df, _ = ext.linear_fwd(dy, wt, None, "", None, False)
        return ext.act_bwd(df, z, "gelu")

    def plain8():
        df, _ = ext.linear_fwd(dy, wt, None, "", None, False)
        dz = ext.act_bwd(df, z, "gelu")
        return ext.fp8_cast(dz, s8, am)

    for name, fn in (("fused", fused), ("fused+e4m3", fused8),
                     ("plain+actbwd", plain), ("plain+actbwd+cast", plain8)):
        ms = t_ms(fn)
        print(f"gradact {name:18s} {ms:7.3f} ms  {tf / (ms * 1e-3):6.0f} TF/s")


def dw():
    ext = _backend.ext()
    print(f"splitm env = {os.environ.get('JIMM_AMD_DW_SPLITM', '(default)')}")
    tot = 0.0
    for n, k in [(3072, 1024), (1024, 1024), (4096, 1024), (1024, 4096)]:
        dz = torch.randn(M, n, device="cuda").bfloat16()
        x = torch.randn(M, k, device="cuda").bfloat16()
        ms = t_ms(lambda: ext.gemm_tn_8p(dz, x))
        tot += ms
        print(f"dw ({n},{k}) {ms:7.3f} ms  {2.0 * M * n * k / 1e12 / (ms * 1e-3):6.0f} TF/s")
    print(f"dw total {tot:.3f} ms")


def ln():
    ext = _backend.ext()
    x = torch.randn(M, 1024, device="cuda").bfloat16()
    w = torch.randn(1024, device="cuda").bfloat16()
    b = torch.randn(1024, device="cuda").bfloat16()
    s8 = torch.ones(1, device="cuda")
    am = torch.zeros(1, device="cuda")
    print(f"ln_fwd      {t_ms(lambda: ext.layernorm_fwd(x, w, b, 1e-6)):7.3f} ms")
    print(f"ln_fwd_fp8  {t_ms(lambda: ext.layernorm_fwd_fp8(x, w, b, 1e-6, s8, am)):7.3f} ms")


if __name__ == "__main__":
    which = sys.argv[1] if len(sys.argv) > 1 else "all"
    if which in ("gradact", "all"):
        gradact()
    if which in ("dw", "all"):
        dw()
    if which in ("ln", "all"):
        ln()
