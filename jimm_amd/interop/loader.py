"""Checkpoint IO: resolve local dir / local file / HF-Hub repo id to
(state_dict, config) and write HF-compatible safetensors back.

Reference behavior reproduced: ``load_params_and_config``
(/root/reference/src/jimm/common/utils.py:28-107):
  * local directory -> config.json + model.safetensors | pytorch_model.bin
  * local single file -> weights only (config=None -> caller shape-infers)
  * otherwise -> HF Hub download (config fetch failure tolerated, weights
    failure fatal — utils.py:93-105); `model/` subdir fallback (utils.py:77-86)
  * ``use_pytorch`` selects pytorch_model.bin via torch.load(map_location=cpu)
    (utils.py:56,70-71)

Plus the save path the reference lacks (SURVEY §5 checkpoint/resume).
"""

from __future__ import annotations

import json
import os
from typing import Any

import torch

SAFETENSORS_NAME = "model.safetensors"
PYTORCH_NAME = "pytorch_model.bin"
CONFIG_NAME = "config.json"

# HF buffer keys that are never parameters (rotary/position id caches)
IGNORE_PATTERNS = (".position_ids",)


def _load_file_weights(path: str) -> dict[str, torch.Tensor]:
    if path.endswith(".safetensors"):
        from safetensors.torch import load_file

        return load_file(path)
    return torch.load(path, map_location="cpu", weights_only=True)


def load_checkpoint(
    model_name_or_path: str,
    *,
    use_pytorch: bool = False,
) -> tuple[dict[str, torch.Tensor], dict[str, Any] | None]:
    """Returns (hf_state_dict on cpu, config dict or None)."""
    weights_name = PYTORCH_NAME if use_pytorch else SAFETENSORS_NAME

    if os.path.isdir(model_name_or_path):
        d = model_name_or_path
        config = None
        cfg_path = os.path.join(d, CONFIG_NAME)
        if not os.path.exists(cfg_path) and os.path.exists(os.path.join(d, "model", CONFIG_NAME)):
            cfg_path = os.path.join(d, "model", CONFIG_NAME)  # utils.py:77-86 subdir fallback
        if os.path.exists(cfg_path):
            with open(cfg_path) as f:
                config = json.load(f)
        w_path = os.path.join(d, weights_name)
        if not os.path.exists(w_path):
            # accept either format if the requested one is absent
            for alt in (SAFETENSORS_NAME, PYTORCH_NAME):
                if os.path.exists(os.path.join(d, alt)):
                    w_path = os.path.join(d, alt)
                    break
        if not os.path.exists(w_path):
            raise FileNotFoundError(f"no weights ({weights_name}) found under {d}")  # utils.py:104-105
        return _load_file_weights(w_path), config

    if os.path.isfile(model_name_or_path):
        return _load_file_weights(model_name_or_path), None

    # HF Hub repo id
    try:
        from huggingface_hub import hf_hub_download
    except ImportError as e:
        raise FileNotFoundError(
            f"{model_name_or_path!r} is not a local path and huggingface_hub is unavailable"
        ) from e
    config = None
    try:
        with open(hf_hub_download(model_name_or_path, CONFIG_NAME)) as f:
            config = json.load(f)
    except Exception:
        config = None  # tolerated (utils.py:93-98)
    w_path = hf_hub_download(model_name_or_path, weights_name)  # failure is fatal
    return _load_file_weights(w_path), config


def save_checkpoint(state_dict: dict[str, torch.Tensor], config: dict[str, Any], save_dir: str) -> None:
    from safetensors.torch import save_file

    os.makedirs(save_dir, exist_ok=True)
    with open(os.path.join(save_dir, CONFIG_NAME), "w") as f:
        json.dump(config, f, indent=2)
    cpu_sd = {k: v.detach().contiguous().cpu() for k, v in state_dict.items()}
    save_file(cpu_sd, os.path.join(save_dir, SAFETENSORS_NAME))


class KeyMap:
    """Consumes an HF state dict with full-coverage bookkeeping.

    Mirrors the reference's bidirectional coverage asserts (SURVEY §2.3:
    models/vit.py:252-268): every source key must be consumed (modulo
    IGNORE_PATTERNS), every target parameter assigned, shapes checked.
    """

    def __init__(self, sd: dict[str, torch.Tensor]):
        self.sd = sd
        self.used: set[str] = set()
        self.out: dict[str, torch.Tensor] = {}

    def has(self, key: str) -> bool:
        return key in self.sd

    def take(self, key: str) -> torch.Tensor:
        if key not in self.sd:
            raise KeyError(f"checkpoint missing key {key!r}")
        self.used.add(key)
        return self.sd[key]

    def put(self, target: str, value: torch.Tensor) -> None:
        if target in self.out:
            raise ValueError(f"target key {target!r} assigned twice")
        self.out[target] = value

    def copy(self, target: str, src: str) -> None:
        self.put(target, self.take(src))

    def finish(self, model: torch.nn.Module, *, dtype: torch.dtype | None = None) -> None:
        leftovers = [
            k for k in self.sd if k not in self.used and not any(k.endswith(p) for p in IGNORE_PATTERNS)
        ]
        if leftovers:
            raise ValueError(f"unconsumed checkpoint keys: {leftovers[:10]}{'...' if len(leftovers) > 10 else ''}")
        target_sd = model.state_dict()
        missing = sorted(set(target_sd) - set(self.out))
        extra = sorted(set(self.out) - set(target_sd))
        if missing or extra:
            raise ValueError(f"coverage mismatch: missing={missing[:10]} extra={extra[:10]}")
        for k, v in self.out.items():
            if tuple(v.shape) != tuple(target_sd[k].shape):
                raise ValueError(f"shape mismatch for {k}: checkpoint {tuple(v.shape)} vs model {tuple(target_sd[k].shape)}")
        out = self.out
        if dtype is not None:
            out = {k: (v.to(dtype) if v.is_floating_point() else v) for k, v in out.items()}
        model.load_state_dict(out, strict=True)
