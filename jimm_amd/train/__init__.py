from jimm_amd.train.adam import Adam  # noqa: F401
from jimm_amd.train.data import SyntheticImages, SyntheticImageText  # noqa: F401
from jimm_amd.train.metrics import Meter  # noqa: F401
from jimm_amd.train.trainer import TrainConfig, Trainer, init_distributed  # noqa: F401

__all__ = ["Adam", "SyntheticImages", "SyntheticImageText", "Meter", "TrainConfig", "Trainer", "init_distributed"]
