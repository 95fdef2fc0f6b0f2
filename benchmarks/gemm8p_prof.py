"""Run ONE GEMM engine repeatedly (for rocprofv3 PMC collection).

env: SHAPE="M,N,K" (default 4096,4096,4096), ENGINE=8p|2ph|blas, ITERS.
"""

import os, sys, time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

import jimm_amd  # noqa: F401
from jimm_amd.ops import _backend

M, N, K = (int(v) for v in os.environ.get("SHAPE", "4096,4096,4096").split(","))
engine = os.environ.get("ENGINE", "8p")
iters = int(os.environ.get("ITERS", "10"))

ext = _backend.ext()
dev = torch.device("cuda:0")
torch.manual_seed(0)
x = (torch.rand(M, K, device=dev) * 2 - 1).bfloat16()
w = ((torch.rand(N, K, device=dev) * 2 - 1) / K**0.5).bfloat16()
b = torch.randn(N, device=dev).bfloat16()

if engine == "2ph":
    os.environ["JIMM_AMD_GEMM_TILE"] = "256"
fn = (lambda: torch.nn.functional.linear(x, w, b)) if engine == "blas" else (
    lambda: ext.linear_fwd(x, w, b, "", None, False))

for _ in range(3):
    fn()
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(iters):
    fn()
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / iters
print(f"ENGINE={engine} SWZ={os.environ.get('JIMM_AMD_GEMM_SWZ','1')} "
      f"M={M} N={N} K={K}: {2*M*N*K/1e12/dt:.1f} TF/s ({dt*1e3:.3f} ms)", flush=True)
