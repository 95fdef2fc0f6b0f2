from jimm_amd.models.common.transformer import Encoder, EncoderBlock  # noqa: F401
from jimm_amd.models.common.text import TextTransformer  # noqa: F401
from jimm_amd.models.common.vit import MAPHead, VisionTransformerBase  # noqa: F401

__all__ = ["Encoder", "EncoderBlock", "TextTransformer", "MAPHead", "VisionTransformerBase"]
