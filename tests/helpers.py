"""Shared test fixtures: safetensors writing (HF layout, the layout cake's
VarBuilder reads — utils/mod.rs:251-370) and golden fixture loading."""
import json
import os

import numpy as np

from oracle import Config, LayerWeights, ModelWeights


def flatten(w: ModelWeights, cfg: Config) -> dict:
    out = {"model.embed_tokens.weight": w.embed_tokens,
           "model.norm.weight": w.norm}
    if not cfg.tie_word_embeddings:
        out["lm_head.weight"] = w.lm_head
    for i, lw in enumerate(w.layers):
        p = f"model.layers.{i}."
        out[p + "input_layernorm.weight"] = lw.input_layernorm
        out[p + "post_attention_layernorm.weight"] = lw.post_attention_layernorm
        out[p + "self_attn.q_proj.weight"] = lw.q_proj
        out[p + "self_attn.k_proj.weight"] = lw.k_proj
        out[p + "self_attn.v_proj.weight"] = lw.v_proj
        out[p + "self_attn.o_proj.weight"] = lw.o_proj
        out[p + "mlp.gate_proj.weight"] = lw.gate_proj
        out[p + "mlp.up_proj.weight"] = lw.up_proj
        out[p + "mlp.down_proj.weight"] = lw.down_proj
        if lw.q_norm is not None:
            out[p + "self_attn.q_norm.weight"] = lw.q_norm
            out[p + "self_attn.k_norm.weight"] = lw.k_norm
    return out


def weights_to_safetensors(w: ModelWeights, cfg: Config, path: str):
    from safetensors.numpy import save_file
    tensors = {k: np.ascontiguousarray(v, dtype=np.float32)
               for k, v in flatten(w, cfg).items()}
    save_file(tensors, path)


def fixture_weights(golden_dir: str, name: str):
    """Rebuild (cfg_json, Config, ModelWeights, npz) from a committed golden
    fixture."""
    z = np.load(os.path.join(golden_dir, f"{name}.npz"))
    with open(os.path.join(golden_dir, f"{name}.config.json")) as f:
        cfg_json = json.load(f)
    cfg = Config.from_json(cfg_json)
    t = {k[2:]: z[k] for k in z.files if k.startswith("w.")}
    layers = []
    for i in range(cfg.num_hidden_layers):
        p = f"model.layers.{i}."
        layers.append(LayerWeights(
            input_layernorm=t[p + "input_layernorm.weight"],
            post_attention_layernorm=t[p + "post_attention_layernorm.weight"],
            q_proj=t[p + "self_attn.q_proj.weight"],
            k_proj=t[p + "self_attn.k_proj.weight"],
            v_proj=t[p + "self_attn.v_proj.weight"],
            o_proj=t[p + "self_attn.o_proj.weight"],
            gate_proj=t[p + "mlp.gate_proj.weight"],
            up_proj=t[p + "mlp.up_proj.weight"],
            down_proj=t[p + "mlp.down_proj.weight"],
            q_norm=t.get(p + "self_attn.q_norm.weight"),
            k_norm=t.get(p + "self_attn.k_norm.weight"),
        ))
    embed = t["model.embed_tokens.weight"]
    w = ModelWeights(
        embed_tokens=embed, norm=t["model.norm.weight"],
        lm_head=embed if cfg.tie_word_embeddings else t["lm_head.weight"],
        layers=layers)
    return cfg_json, cfg, w, z


def quantize_bf16(a: np.ndarray) -> np.ndarray:
    """Round f32 -> bf16 -> f32 (RNE), matching the device conversion."""
    u = np.ascontiguousarray(a, dtype=np.float32).view(np.uint32)
    r = u + 0x7FFF + ((u >> 16) & 1)
    nan = (u & 0x7FFFFFFF) > 0x7F800000
    r = np.where(nan, np.uint32(0x7FC00000), r & 0xFFFF0000)
    return r.astype(np.uint32).view(np.float32).reshape(a.shape)
