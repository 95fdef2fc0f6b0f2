"""HF checkpoint interop for SigLIP.

Reference mapping: /root/reference/src/jimm/models/siglip.py:176-385.
The HF MAP head stores torch's fused ``in_proj_weight/in_proj_bias``; our
MAPHead uses the same fused layout natively so no split/re-fuse is needed
(the reference splits into thirds at siglip.py:352-363).
``logit_scale``/``logit_bias`` squeeze from shape (1,) (siglip.py:322-323).
"""

from __future__ import annotations

import math

import torch

from jimm_amd.interop.clip_hf import _map_tower_layers, _unmap_tower_layers
from jimm_amd.interop.loader import KeyMap, load_checkpoint, save_checkpoint


def _parse_config(cfg: dict) -> dict:
    tc, vc = cfg["text_config"], cfg["vision_config"]
    return dict(
        image_resolution=vc.get("image_size", 256),
        vision_layers=vc["num_hidden_layers"],
        vision_width=vc["hidden_size"],
        vision_patch_size=vc.get("patch_size", 16),
        vision_mlp_dim=vc.get("intermediate_size"),
        vision_heads=vc.get("num_attention_heads"),
        context_length=tc.get("max_position_embeddings", 64),
        vocab_size=tc.get("vocab_size", 32000),
        transformer_width=tc["hidden_size"],
        transformer_heads=tc["num_attention_heads"],
        transformer_layers=tc["num_hidden_layers"],
        transformer_mlp_dim=tc.get("intermediate_size"),
    )


def _infer_config(sd: dict[str, torch.Tensor], image_size: int | None = None) -> dict:
    """Shape inference (reference siglip.py:193-207 — config only for image_size)."""
    conv_w = sd["vision_model.embeddings.patch_embedding.weight"]
    vision_width = conv_w.shape[0]
    # v1: conv (H, 3, P, P); v2: linear (H, P*P*3) over channels-last patches
    patch = conv_w.shape[-1] if conv_w.dim() == 4 else int(math.isqrt(conv_w.shape[1] // 3))
    v_pos = sd["vision_model.embeddings.position_embedding.weight"].shape[0]
    img = image_size or int(math.isqrt(v_pos)) * patch
    n_layers = lambda pref: 1 + max(int(k.split(".")[3]) for k in sd if k.startswith(pref + ".encoder.layers."))
    tw = sd["text_model.embeddings.token_embedding.weight"].shape[1]
    return dict(
        image_resolution=img,
        vision_layers=n_layers("vision_model"),
        vision_width=vision_width,
        vision_patch_size=patch,
        vision_mlp_dim=sd["vision_model.encoder.layers.0.mlp.fc1.weight"].shape[0],
        context_length=sd["text_model.embeddings.position_embedding.weight"].shape[0],
        vocab_size=sd["text_model.embeddings.token_embedding.weight"].shape[0],
        transformer_width=tw,
        transformer_heads=max(1, tw // 64),  # siglip.py:217
        transformer_layers=n_layers("text_model"),
        transformer_mlp_dim=sd["text_model.encoder.layers.0.mlp.fc1.weight"].shape[0],
    )


def map_siglip(sd: dict[str, torch.Tensor], vision_layers: int, text_layers: int) -> KeyMap:
    m = KeyMap(sd)
    m.put("logit_scale", m.take("logit_scale").reshape(()))
    m.put("logit_bias", m.take("logit_bias").reshape(()))
    m.copy("text_projection.weight", "text_model.head.weight")
    m.copy("text_projection.bias", "text_model.head.bias")
    # vision tower (no CLS token, no pre-norm). SigLIP2 stores the patch
    # embedding as a Linear over channels-LAST flattened patches
    # (H, P*P*C) — transformers' Siglip2 image processor packs each patch
    # as image[py, px, c]; reshape to the conv layout (H, C, P, P).
    pw = m.take("vision_model.embeddings.patch_embedding.weight")
    if pw.dim() == 2:
        H_, feat = pw.shape
        P = int(math.isqrt(feat // 3))
        pw = pw.view(H_, P, P, 3).permute(0, 3, 1, 2).contiguous()
    m.put("vision_model.patch_weight", pw)
    m.copy("vision_model.patch_bias", "vision_model.embeddings.patch_embedding.bias")
    m.put("vision_model.pos_embedding", m.take("vision_model.embeddings.position_embedding.weight").unsqueeze(0))
    m.copy("vision_model.ln_post.weight", "vision_model.post_layernorm.weight")
    m.copy("vision_model.ln_post.bias", "vision_model.post_layernorm.bias")
    _map_tower_layers(m, "vision_model", "vision_model", vision_layers)
    # MAP head (fused in_proj both sides)
    m.copy("vision_model.map_head.probe", "vision_model.head.probe")
    m.copy("vision_model.map_head.in_proj_weight", "vision_model.head.attention.in_proj_weight")
    m.copy("vision_model.map_head.in_proj_bias", "vision_model.head.attention.in_proj_bias")
    m.copy("vision_model.map_head.out_proj.weight", "vision_model.head.attention.out_proj.weight")
    m.copy("vision_model.map_head.out_proj.bias", "vision_model.head.attention.out_proj.bias")
    m.copy("vision_model.map_head.layernorm.weight", "vision_model.head.layernorm.weight")
    m.copy("vision_model.map_head.layernorm.bias", "vision_model.head.layernorm.bias")
    m.copy("vision_model.map_head.fc1.weight", "vision_model.head.mlp.fc1.weight")
    m.copy("vision_model.map_head.fc1.bias", "vision_model.head.mlp.fc1.bias")
    m.copy("vision_model.map_head.fc2.weight", "vision_model.head.mlp.fc2.weight")
    m.copy("vision_model.map_head.fc2.bias", "vision_model.head.mlp.fc2.bias")
    # text tower
    m.copy("text_model.token_embedding.weight", "text_model.embeddings.token_embedding.weight")
    m.put("text_model.pos_embedding", m.take("text_model.embeddings.position_embedding.weight").unsqueeze(0))
    m.copy("text_model.ln_final.weight", "text_model.final_layer_norm.weight")
    m.copy("text_model.ln_final.bias", "text_model.final_layer_norm.bias")
    _map_tower_layers(m, "text_model", "text_model", text_layers)
    return m


def load_siglip(cls, model_name_or_path: str, *, use_pytorch: bool = False, dtype: torch.dtype = torch.float32, device="cpu"):
    sd, cfg = load_checkpoint(model_name_or_path, use_pytorch=use_pytorch)
    if cfg:
        kwargs = _parse_config(cfg)
    else:
        kwargs = _infer_config(sd)
    model = cls(**kwargs)
    m = map_siglip(sd, len(model.vision_model.encoder.layers), len(model.text_model.encoder.layers))
    m.finish(model, dtype=dtype)
    return model.to(device)


def save_siglip(model, save_dir: str) -> None:
    sd = model.state_dict()
    out: dict[str, torch.Tensor] = {}
    out["logit_scale"] = sd["logit_scale"].reshape(1)
    out["logit_bias"] = sd["logit_bias"].reshape(1)
    out["text_model.head.weight"] = sd["text_projection.weight"]
    out["text_model.head.bias"] = sd["text_projection.bias"]
    out["vision_model.embeddings.patch_embedding.weight"] = sd["vision_model.patch_weight"]
    out["vision_model.embeddings.patch_embedding.bias"] = sd["vision_model.patch_bias"]
    out["vision_model.embeddings.position_embedding.weight"] = sd["vision_model.pos_embedding"].squeeze(0)
    out["vision_model.post_layernorm.weight"] = sd["vision_model.ln_post.weight"]
    out["vision_model.post_layernorm.bias"] = sd["vision_model.ln_post.bias"]
    out["vision_model.head.probe"] = sd["vision_model.map_head.probe"]
    out["vision_model.head.attention.in_proj_weight"] = sd["vision_model.map_head.in_proj_weight"]
    out["vision_model.head.attention.in_proj_bias"] = sd["vision_model.map_head.in_proj_bias"]
    out["vision_model.head.attention.out_proj.weight"] = sd["vision_model.map_head.out_proj.weight"]
    out["vision_model.head.attention.out_proj.bias"] = sd["vision_model.map_head.out_proj.bias"]
    out["vision_model.head.layernorm.weight"] = sd["vision_model.map_head.layernorm.weight"]
    out["vision_model.head.layernorm.bias"] = sd["vision_model.map_head.layernorm.bias"]
    out["vision_model.head.mlp.fc1.weight"] = sd["vision_model.map_head.fc1.weight"]
    out["vision_model.head.mlp.fc1.bias"] = sd["vision_model.map_head.fc1.bias"]
    out["vision_model.head.mlp.fc2.weight"] = sd["vision_model.map_head.fc2.weight"]
    out["vision_model.head.mlp.fc2.bias"] = sd["vision_model.map_head.fc2.bias"]
    out["text_model.embeddings.token_embedding.weight"] = sd["text_model.token_embedding.weight"]
    out["text_model.embeddings.position_embedding.weight"] = sd["text_model.pos_embedding"].squeeze(0)
    out["text_model.final_layer_norm.weight"] = sd["text_model.ln_final.weight"]
    out["text_model.final_layer_norm.bias"] = sd["text_model.ln_final.bias"]
    nv, nt = len(model.vision_model.encoder.layers), len(model.text_model.encoder.layers)
    _unmap_tower_layers(sd, out, "vision_model", "vision_model", nv)
    _unmap_tower_layers(sd, out, "text_model", "text_model", nt)
    vb, tb = model.vision_model, model.text_model
    cfg = {
        "model_type": "siglip",
        "architectures": ["SiglipModel"],
        "text_config": {
            "model_type": "siglip_text_model",
            "hidden_size": tb.token_embedding.embedding_dim,
            "intermediate_size": tb.encoder.layers[0].fc1.out_features,
            "num_hidden_layers": nt,
            "num_attention_heads": tb.encoder.layers[0].num_heads,
            "max_position_embeddings": tb.pos_embedding.shape[1],
            "vocab_size": tb.token_embedding.num_embeddings,
            "hidden_act": "gelu_pytorch_tanh",
        },
        "vision_config": {
            "model_type": "siglip_vision_model",
            "hidden_size": vb.hidden_size,
            "intermediate_size": vb.encoder.layers[0].fc1.out_features,
            "num_hidden_layers": nv,
            "num_attention_heads": vb.encoder.layers[0].num_heads,
            "image_size": vb.img_size,
            "patch_size": vb.patch_size,
            "hidden_act": "gelu_pytorch_tanh",
        },
    }
    save_checkpoint(out, cfg, save_dir)
