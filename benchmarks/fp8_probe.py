"""Probe torch._scaled_mm (hipBLASLt fp8) on gfx950: availability + TF/s."""
import time
import torch

def bench(fn, iters=20, warmup=5):
    for _ in range(warmup): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters

dev = torch.device("cuda:0")
print("fp8 dtypes:", [d for d in (getattr(torch, n, None) for n in
      ("float8_e4m3fn", "float8_e4m3fnuz", "float8_e5m2")) if d])
for M, N, K in [(50432, 2304, 768), (50432, 3072, 768), (50432, 768, 3072), (8192, 8192, 8192)]:
    x = torch.randn(M, K, device=dev)
    w = torch.randn(N, K, device=dev)
    sx = x.abs().max() / 448.0
    sw = w.abs().max() / 448.0
    try:
        x8 = (x / sx).to(torch.float8_e4m3fn)
        w8 = (w / sw).to(torch.float8_e4m3fn)
        out = torch._scaled_mm(x8, w8.t(), scale_a=sx.view(1, 1), scale_b=sw.view(1, 1), out_dtype=torch.bfloat16)
        t = bench(lambda: torch._scaled_mm(x8, w8.t(), scale_a=sx.view(1, 1), scale_b=sw.view(1, 1), out_dtype=torch.bfloat16))
        ref = x @ w.t()
        err = (out.float() - ref).abs().max() / ref.abs().max()
        print(f"M={M} N={N} K={K}: {2*M*N*K/1e12/t:7.1f} TF/s ({t*1e3:.2f} ms) relerr {err:.3f}")
    except Exception as e:
        print(f"M={M} N={N} K={K}: FAILED {e!r}")
