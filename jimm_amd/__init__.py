"""jimm_amd — MI355X-native image / image-text model library.

A from-scratch, MI355X-first framework with the capabilities of the reference
library ``pythoncrazy/jimm`` (JAX/Flax): ViT, CLIP and SigLIP
model families with HuggingFace checkpoint interop, built on PyTorch-ROCm with
hand-written HIP/CDNA4 kernels (MFMA bf16, LDS-staged tiles) for the hot ops
and RCCL (``torch.distributed`` backend "nccl") over xGMI for data-parallel
training and the cross-GPU contrastive/sigmoid losses.

Layer map (mirrors SURVEY.md §1, re-designed for MI355X):
  ops/       — op surface: HIP kernels on GPU, PyTorch reference on CPU
  models/    — ViT, CLIP, SigLIP (public API) over models/common building blocks
  interop/   — HF checkpoint load/save (safetensors + pytorch_model.bin)
  parallel/  — explicit RCCL collectives: bucketed DP grad all-reduce,
               all-gather-with-grad for global-batch contrastive losses
  train/     — trainer, synthetic data pipeline, metrics
"""

__version__ = "0.1.0"

from jimm_amd.models import CLIP, SigLIP, VisionTransformer  # noqa: F401

__all__ = ["VisionTransformer", "CLIP", "SigLIP", "__version__"]
