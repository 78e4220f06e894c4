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


def random_fp8_model(cfg: Config, seed: int = 7):
    """Synthetic fp8 (e4m3fn + blockwise scale_inv) model: returns
    (tensors dict for safetensors, dequantized-f32 ModelWeights for the
    oracle).  Restates utils/fp8.rs:42-64 blockwise semantics."""
    from oracle import fp8_dequant
    rng = np.random.default_rng(seed)
    H, I, V = cfg.hidden_size, cfg.intermediate_size, cfg.vocab_size
    hd, nh, nkv = cfg.hd, cfg.num_attention_heads, cfg.num_key_value_heads

    def w8(n, k):
        b = rng.integers(0, 256, size=(n, k)).astype(np.uint8)
        b[(b & 0x7F) == 0x7F] ^= 0x08  # no NaN encodings
        sc = (3e-4 + 2e-4 * rng.random(((n + 127) // 128, (k + 127) // 128))
              ).astype(np.float32)
        return b, sc

    def f32(*shape):
        return (rng.standard_normal(shape) * 0.02).astype(np.float32)

    tensors = {}
    layers = []
    for i in range(cfg.num_hidden_layers):
        p = f"model.layers.{i}."
        parts = {}
        for name, n, k in (("self_attn.q_proj", nh * hd, H),
                           ("self_attn.k_proj", nkv * hd, H),
                           ("self_attn.v_proj", nkv * hd, H),
                           ("self_attn.o_proj", H, nh * hd),
                           ("mlp.gate_proj", I, H),
                           ("mlp.up_proj", I, H),
                           ("mlp.down_proj", H, I)):
            b, sc = w8(n, k)
            tensors[p + name + ".weight"] = b
            tensors[p + name + ".weight_scale_inv"] = sc
            parts[name] = fp8_dequant(b, sc)
        ln1, ln2 = 1.0 + f32(H), 1.0 + f32(H)
        tensors[p + "input_layernorm.weight"] = ln1
        tensors[p + "post_attention_layernorm.weight"] = ln2
        layers.append(LayerWeights(
            input_layernorm=ln1, post_attention_layernorm=ln2,
            q_proj=parts["self_attn.q_proj"], k_proj=parts["self_attn.k_proj"],
            v_proj=parts["self_attn.v_proj"], o_proj=parts["self_attn.o_proj"],
            gate_proj=parts["mlp.gate_proj"], up_proj=parts["mlp.up_proj"],
            down_proj=parts["mlp.down_proj"]))
    embed = f32(V, H)
    norm = 1.0 + f32(H)
    head = embed if cfg.tie_word_embeddings else f32(V, H)
    tensors["model.embed_tokens.weight"] = embed
    tensors["model.norm.weight"] = norm
    if not cfg.tie_word_embeddings:
        tensors["lm_head.weight"] = head
    w = ModelWeights(embed_tokens=embed, norm=norm, lm_head=head,
                     layers=layers)
    return tensors, w


def save_safetensors_raw(tensors: dict, path: str):
    from safetensors.numpy import save_file
    save_file({k: np.ascontiguousarray(v) for k, v in tensors.items()}, path)
