from jimm_amd.ops.functional import (  # noqa: F401
    add_cls_pos,
    attention,
    attention_qkv,
    layer_norm,
    linear,
    patch_embed,
    quickgelu,
)
from jimm_amd.ops._backend import has_ext  # noqa: F401

__all__ = [
    "add_cls_pos",
    "attention",
    "attention_qkv",
    "layer_norm",
    "linear",
    "patch_embed",
    "quickgelu",
    "has_ext",
]
