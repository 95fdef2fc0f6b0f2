from jimm_amd.parallel.ddp import DataParallelGrads  # noqa: F401
from jimm_amd.parallel.gather import all_gather_with_grad  # noqa: F401

__all__ = ["DataParallelGrads", "all_gather_with_grad"]
