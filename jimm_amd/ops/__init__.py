from jimm_amd.ops.functional import (  # noqa: F401
    add_cls_pos,
    attention,
    attention_qkv,
    embed_pos,
    layer_norm,
    linear,
    patch_embed,
    quickgelu,
)
from jimm_amd.ops._backend import has_ext  # noqa: F401
from jimm_amd.ops.hip_linear import fp8_enabled, set_fp8  # noqa: F401

__all__ = [
    "fp8_enabled",
    "set_fp8",
    "add_cls_pos",
    "attention",
    "attention_qkv",
    "embed_pos",
    "layer_norm",
    "linear",
    "patch_embed",
    "quickgelu",
    "has_ext",
]
