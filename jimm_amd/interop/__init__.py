from jimm_amd.interop.loader import load_checkpoint, save_checkpoint  # noqa: F401

__all__ = ["load_checkpoint", "save_checkpoint"]
