"""A/B the config-5 (ViT-L/16@384, M=128*577) block GEMM sites:
in-house bf16 MFMA GEMM (gemm8p) vs hipBLASLt e4m3 scaled_mm.

Pure-GEMM timings (inputs pre-quantized) bound what producer-fused fp8 can
save; the +quant column adds the per-call weight quantization the block
path actually pays. Run on the GPU box:
  python benchmarks/fp8_site_ab.py
"""

import torch

from jimm_amd.ops import _backend

M = 128 * 577
SITES = [
    ("qkv  ", M, 3072, 1024),
    ("proj ", M, 1024, 1024),
    ("fc1  ", M, 4096, 1024),
    ("fc2  ", M, 1024, 4096),
    # dX shapes (NN): dz (M,N) @ W (N,K) -> run as NT on W^T copy
    ("dx-qkv", M, 1024, 3072),
    ("dx-fc1", M, 1024, 4096),
    ("dx-fc2", M, 4096, 1024),
]


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
    ext = _backend.ext()
    dev = "cuda"
    print(f"{'site':7s} {'bf16-hip':>9s} {'fp8-pure':>9s} {'fp8+wq':>9s}  (TF/s; M={M})")
    for name, m, n, k in SITES:
        x = torch.randn(m, k, device=dev).bfloat16()
        w = (torch.randn(n, k, device=dev) / 30).bfloat16()
        flops = 2.0 * m * n * k

        def hip():
            y, _ = ext.linear_fwd(x, w, None, "", None, False)
            return y

        x8 = (x.float() / 4).to(torch.float8_e4m3fn)
        w8 = (w.float() * 30).to(torch.float8_e4m3fn)
        sx = torch.ones(1, 1, device=dev)
        sw = torch.ones(1, 1, device=dev)

        def fp8_pure():
            return torch._scaled_mm(x8, w8.t(), scale_a=sx, scale_b=sw,
                                    out_dtype=torch.bfloat16)

        def fp8_wq():
            s = (w.abs().amax().float() / 448.0).clamp(min=1e-12)
            wq = (w * s.reciprocal().to(w.dtype)).to(torch.float8_e4m3fn)
            return torch._scaled_mm(x8, wq.t(), scale_a=sx, scale_b=s.view(1, 1),
                                    out_dtype=torch.bfloat16)

        r = []
        for fn in (hip, fp8_pure, fp8_wq):
            try:
                r.append(flops / (t_ms(fn) * 1e-3) / 1e12)
            except Exception as ex:  # noqa: BLE001
                r.append(float("nan"))
                print("  !", name, type(ex).__name__, str(ex)[:80])
        print(f"{name:7s} {r[0]:9.0f} {r[1]:9.0f} {r[2]:9.0f}")


if __name__ == "__main__":
    main()
