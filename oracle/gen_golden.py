"""Pin the oracle against HF transformers (cake's own semantic ground truth).

cake cannot be compiled in this container (no Rust toolchain/network) and its
hot-path numerics live in the un-vendored candle 0.9.2 dep, which tracks HF
transformers (cache.rs:93-94 cites HF modeling_llama.py).  This script runs HF
transformers 5.15 on torch-2.10 CPU (present in this image) on tiny
random-init Llama-3-style and Qwen3-style configs, copies the SAME weights
into the oracle, checks logits agreement, and commits golden fixtures
(weights + prompt + logits + greedy token ids) under tests/golden/.

Run from the repo root:  python -m oracle.gen_golden
(The GPU box never runs this — it reads the committed .npz fixtures.)
"""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from oracle import Config, LayerWeights, ModelWeights, OracleModel  # noqa: E402

GOLDEN_DIR = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "tests", "golden")

SEED = 299792458  # cake's default seed (cake-cli lib.rs:180)


def tiny_llama_cfg():
    """Llama-3 shaped (GQA, untied head, llama3 rope scaling), tiny dims."""
    return dict(
        model_type="llama", hidden_size=64, intermediate_size=128,
        vocab_size=256, num_hidden_layers=2, num_attention_heads=4,
        num_key_value_heads=2, head_dim=16, rms_norm_eps=1e-5,
        rope_theta=500000.0, max_position_embeddings=256,
        tie_word_embeddings=False,
        rope_scaling=dict(rope_type="llama3", factor=8.0,
                          low_freq_factor=1.0, high_freq_factor=4.0,
                          original_max_position_embeddings=64),
    )


def tiny_mistral_cfg():
    """Mistral shaped (sliding-window attention, plain rope), tiny dims.
    The prompt (48 tokens) exceeds the window (24) so the windowed mask is
    actually exercised (cache.rs:173-205 trim semantics)."""
    return dict(
        model_type="mistral", hidden_size=64, intermediate_size=128,
        vocab_size=256, num_hidden_layers=2, num_attention_heads=4,
        num_key_value_heads=2, head_dim=16, rms_norm_eps=1e-5,
        rope_theta=10000.0, max_position_embeddings=256,
        tie_word_embeddings=False, sliding_window=24,
    )


def tiny_qwen3_cfg():
    """Qwen3 shaped (qk_norm, tied embeddings), tiny dims."""
    return dict(
        model_type="qwen3", hidden_size=64, intermediate_size=128,
        vocab_size=256, num_hidden_layers=2, num_attention_heads=4,
        num_key_value_heads=2, head_dim=16, rms_norm_eps=1e-6,
        rope_theta=1000000.0, max_position_embeddings=256,
        tie_word_embeddings=True,
    )


def tiny_qwen3_swa_cfg():
    """Qwen3 with INTERLEAVED sliding-window layers (max_window_layers=1:
    layer 0 full, layer 1 windowed) — the per-layer window pattern
    Gemma3/EXAONE-class models use."""
    cfg = tiny_qwen3_cfg()
    cfg.update(use_sliding_window=True, sliding_window=24,
               max_window_layers=1)
    return cfg


def hf_model(cfg_json):
    import torch
    import transformers
    torch.manual_seed(SEED)
    if cfg_json["model_type"] == "llama":
        cfg = transformers.LlamaConfig(
            hidden_size=cfg_json["hidden_size"],
            intermediate_size=cfg_json["intermediate_size"],
            vocab_size=cfg_json["vocab_size"],
            num_hidden_layers=cfg_json["num_hidden_layers"],
            num_attention_heads=cfg_json["num_attention_heads"],
            num_key_value_heads=cfg_json["num_key_value_heads"],
            head_dim=cfg_json["head_dim"],
            rms_norm_eps=cfg_json["rms_norm_eps"],
            rope_theta=cfg_json["rope_theta"],
            max_position_embeddings=cfg_json["max_position_embeddings"],
            tie_word_embeddings=cfg_json["tie_word_embeddings"],
            rope_scaling=cfg_json.get("rope_scaling"),
            attention_bias=False, mlp_bias=False,
        )
        m = transformers.LlamaForCausalLM(cfg)
    elif cfg_json["model_type"] == "mistral":
        cfg = transformers.MistralConfig(
            hidden_size=cfg_json["hidden_size"],
            intermediate_size=cfg_json["intermediate_size"],
            vocab_size=cfg_json["vocab_size"],
            num_hidden_layers=cfg_json["num_hidden_layers"],
            num_attention_heads=cfg_json["num_attention_heads"],
            num_key_value_heads=cfg_json["num_key_value_heads"],
            head_dim=cfg_json["head_dim"],
            rms_norm_eps=cfg_json["rms_norm_eps"],
            rope_theta=cfg_json["rope_theta"],
            max_position_embeddings=cfg_json["max_position_embeddings"],
            tie_word_embeddings=cfg_json["tie_word_embeddings"],
            sliding_window=cfg_json["sliding_window"],
            attn_implementation="eager",
        )
        m = transformers.MistralForCausalLM(cfg)
    else:
        cfg = transformers.Qwen3Config(
            hidden_size=cfg_json["hidden_size"],
            intermediate_size=cfg_json["intermediate_size"],
            vocab_size=cfg_json["vocab_size"],
            num_hidden_layers=cfg_json["num_hidden_layers"],
            num_attention_heads=cfg_json["num_attention_heads"],
            num_key_value_heads=cfg_json["num_key_value_heads"],
            head_dim=cfg_json["head_dim"],
            rms_norm_eps=cfg_json["rms_norm_eps"],
            rope_theta=cfg_json["rope_theta"],
            max_position_embeddings=cfg_json["max_position_embeddings"],
            tie_word_embeddings=cfg_json["tie_word_embeddings"],
        )
        if cfg_json.get("use_sliding_window"):
            cfg.use_sliding_window = True
            cfg.sliding_window = cfg_json["sliding_window"]
            cfg.max_window_layers = cfg_json["max_window_layers"]
            cfg.layer_types = ["full_attention" if i <
                               cfg_json["max_window_layers"] else
                               "sliding_attention"
                               for i in range(cfg_json["num_hidden_layers"])]
            cfg._attn_implementation = "eager"
        m = transformers.Qwen3ForCausalLM(cfg)
    m.eval()
    m = m.float()
    return m


def hf_to_oracle(m, cfg: Config) -> ModelWeights:
    sd = {k: v.detach().numpy().astype(np.float32)
          for k, v in m.state_dict().items()}
    layers = []
    for i in range(cfg.num_hidden_layers):
        p = f"model.layers.{i}."
        layers.append(LayerWeights(
            input_layernorm=sd[p + "input_layernorm.weight"],
            post_attention_layernorm=sd[p + "post_attention_layernorm.weight"],
            q_proj=sd[p + "self_attn.q_proj.weight"],
            k_proj=sd[p + "self_attn.k_proj.weight"],
            v_proj=sd[p + "self_attn.v_proj.weight"],
            o_proj=sd[p + "self_attn.o_proj.weight"],
            gate_proj=sd[p + "mlp.gate_proj.weight"],
            up_proj=sd[p + "mlp.up_proj.weight"],
            down_proj=sd[p + "mlp.down_proj.weight"],
            q_norm=sd.get(p + "self_attn.q_norm.weight"),
            k_norm=sd.get(p + "self_attn.k_norm.weight"),
        ))
    embed = sd["model.embed_tokens.weight"]
    return ModelWeights(
        embed_tokens=embed,
        norm=sd["model.norm.weight"],
        lm_head=embed if cfg.tie_word_embeddings else sd["lm_head.weight"],
        layers=layers,
    )


def flatten_weights(w: ModelWeights, cfg: Config) -> dict:
    """HF safetensors naming — the layout cake's VarBuilder reads
    (utils/mod.rs:251-370) and the GPU engine's loader reads too."""
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


def run_one(name, cfg_json, prompt_len=17, gen=12):
    import torch
    import json
    m = hf_model(cfg_json)
    cfg = Config.from_json(cfg_json)
    w = hf_to_oracle(m, cfg)
    oracle = OracleModel(cfg, w)

    rng = np.random.default_rng(SEED)
    prompt = rng.integers(0, cfg.vocab_size, size=prompt_len).astype(np.int64)

    # HF prefill logits (last position)
    with torch.no_grad():
        hf_out = m(torch.tensor(prompt[None, :]))
    hf_logits = hf_out.logits[0, -1].numpy()

    # HF greedy generation
    with torch.no_grad():
        hf_gen = m.generate(
            torch.tensor(prompt[None, :]), max_new_tokens=gen,
            do_sample=False, use_cache=True,
            pad_token_id=0)[0, prompt_len:].numpy()

    # Oracle
    or_logits = oracle.forward(prompt[None, :], 0)[0]
    oracle.reset()
    or_gen = np.array(oracle.generate_greedy(list(prompt), gen))

    diff = np.max(np.abs(hf_logits - or_logits))
    rel = diff / max(1e-9, np.max(np.abs(hf_logits)))
    tok_match = bool(np.array_equal(hf_gen, or_gen))
    print(f"[{name}] logits max-abs-diff vs HF = {diff:.3e} (rel {rel:.3e}); "
          f"greedy tokens bit-exact = {tok_match}")
    assert rel < 2e-4, f"{name}: oracle does not match HF transformers"
    assert tok_match, f"{name}: greedy token ids differ from HF"

    os.makedirs(GOLDEN_DIR, exist_ok=True)
    arrs = {"w." + k: v for k, v in flatten_weights(w, cfg).items()}
    arrs.update(prompt=prompt, hf_logits=hf_logits, hf_gen=hf_gen,
                oracle_logits=or_logits)
    np.savez_compressed(os.path.join(GOLDEN_DIR, f"{name}.npz"), **arrs)
    with open(os.path.join(GOLDEN_DIR, f"{name}.config.json"), "w") as f:
        json.dump(cfg_json, f, indent=1)
    print(f"[{name}] fixture written to tests/golden/{name}.npz")


def main():
    run_one("tiny_llama3", tiny_llama_cfg())
    run_one("tiny_qwen3", tiny_qwen3_cfg())
    run_one("tiny_mistral", tiny_mistral_cfg(), prompt_len=48)
    run_one("tiny_qwen3swa", tiny_qwen3_swa_cfg(), prompt_len=48)


if __name__ == "__main__":
    main()
