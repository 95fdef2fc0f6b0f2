"""Microbench: flash attention fwd + fused bwd kernels on model-zoo shapes."""
import math, os, sys, time
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
import jimm_amd  # noqa
from jimm_amd.ops import _backend

def bench(fn, iters=30, warmup=8):
    for _ in range(warmup): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters

ext = _backend.ext()
dev = torch.device("cuda:0")
for B, H, L, causal, tag in [
    (256, 12, 197, False, "ViT-B/16 bs256"),
    (64, 16, 577, False, "ViT-L/16-384 bs64"),
    (256, 8, 77, True, "CLIP text bs256"),
    (256, 12, 257, False, "SigLIP-256 bs256"),
]:
    q = torch.randn(B, H, L, 64, device=dev).bfloat16()
    k = torch.randn_like(q); v = torch.randn_like(q); do = torch.randn_like(q)
    scale = 1 / math.sqrt(64)
    o, lse = ext.attn_fwd(q, k, v, causal, scale)
    oc = o.contiguous()
    dq, dk, dv = torch.empty_like(q), torch.empty_like(k), torch.empty_like(v)
    tf_fwd = 4 * B * H * L * L * 64 / 1e12 * (0.5 if causal else 1.0)
    t_f = bench(lambda: ext.attn_fwd(q, k, v, causal, scale))
    t_b = bench(lambda: ext.attn_bwd_fused(q, k, v, oc, do, lse, dq, dk, dv, causal, scale))
    print(f"{tag:18s} L={L:4d}: fwd {tf_fwd/t_f:6.1f} TF/s ({t_f*1e3:6.3f} ms)  "
          f"bwd {2.5*tf_fwd/t_b:6.1f} TF/s ({t_b*1e3:6.3f} ms)", flush=True)
