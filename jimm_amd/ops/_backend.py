"""HIP extension loading for jimm_amd.

The compiled extension (``jimm_amd/_hip.*.so``) is built IN-TREE by
``setup.py build_ext --inplace`` (or ``__graft_entry__.build()``) with
``PYTORCH_ROCM_ARCH=gfx950`` so the .so travels with the repo snapshot.

Policy (keeps GPU runs honest — no silent eager fallback):
  * on a CUDA/ROCm tensor the HIP kernel MUST run; if the extension is
    missing we raise immediately instead of falling back to eager PyTorch;
  * on CPU tensors the pure-PyTorch reference path runs (it doubles as the
    numerics oracle in tests/).

Set ``JIMM_AMD_FORCE_EAGER=1`` to force the PyTorch path on GPU (debugging
and A/B numerics only; never the default).
"""

from __future__ import annotations


_DEFAULT_TUNABLE = "data/tunableop_mi355x.csv"  # relative to jimm_amd/


def maybe_enable_tunableop() -> bool:
    """Load a TunableOp result table (hipBLASLt algo selections, including
    the split-K picks for the dW shapes — measured 0.23-0.6 -> 0.6-0.9 PF/s).

    Default: the committed jimm_amd/data/tunableop_mi355x.csv (produced by
    benchmarks/bwd_gemm_bench.py --tune on an MI355X). Override the path with
    JIMM_AMD_TUNABLE=<csv>; disable with JIMM_AMD_TUNABLE=0."""
    import os

    path = os.environ.get("JIMM_AMD_TUNABLE", "")
    if path in ("0", "none", "off"):
        return False
    if not path:
        path = os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", _DEFAULT_TUNABLE)
        if not os.path.exists(path):
            return False
    import torch

    if not torch.cuda.is_available():
        return False
    tun = torch.cuda.tunable
    tun.enable(True)
    tun.tuning_enable(False)
    if os.path.exists(path):
        tun.read_file(path)
    return True

import importlib
import os

_ext = None
_ext_err: Exception | None = None


def _load():
    global _ext, _ext_err
    if _ext is not None or _ext_err is not None:
        return _ext
    try:
        _ext = importlib.import_module("jimm_amd._hip")
    except ImportError as e:  # remember why, so the error message is useful
        _ext_err = e
        _ext = None
    return _ext


def has_ext() -> bool:
    return _load() is not None


def ext():
    """Return the HIP extension module, raising loudly if unavailable."""
    m = _load()
    if m is None:
        raise RuntimeError(
            "jimm_amd HIP extension (jimm_amd/_hip) is not built but a GPU "
            "tensor reached a jimm_amd op. Build it in-tree with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            f"Original import error: {_ext_err!r}"
        )
    return m


def force_eager() -> bool:
    return os.environ.get("JIMM_AMD_FORCE_EAGER", "0") == "1"


def use_hip(t) -> bool:
    """True when tensor ``t`` should be handled by the HIP kernels."""
    return t.is_cuda and not force_eager()
