"""Public model API — mirrors /root/reference/src/jimm/models/__init__.py:1-9."""

from jimm_amd.models.clip import CLIP  # noqa: F401
from jimm_amd.models.siglip import SigLIP  # noqa: F401
from jimm_amd.models.vit import VisionTransformer  # noqa: F401

__all__ = ["VisionTransformer", "CLIP", "SigLIP"]
