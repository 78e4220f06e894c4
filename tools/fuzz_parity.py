"""Randomized config fuzz: engine vs oracle across the supported envelope
(head_dim multiples of 8 up to 128, GQA ratios, tied/untied, qk-norm,
sliding windows incl. per-layer, rope scaling).  One-off GPU validation
sweep — deterministic seeds, not part of the pytest suite.

Run: python tools/fuzz_parity.py [n_configs]
"""
import json
import os
import sys
import tempfile

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import cake_amd  # noqa: E402
from oracle import Config, random_weights, OracleModel  # noqa: E402
from tests.helpers import weights_to_safetensors, quantize_bf16  # noqa: E402
from tests.test_gpu_parity import quantized_oracle, rel_err  # noqa: E402


def random_cfg(rng):
    hd = int(rng.choice([8, 16, 24, 32, 40, 48, 64, 96, 128]))
    nh = int(rng.choice([1, 2, 3, 4, 6, 8]))
    divisors = [d for d in range(1, nh + 1) if nh % d == 0]
    nkv = int(rng.choice(divisors))
    layers = int(rng.integers(1, 4))
    qwen = bool(rng.random() < 0.3)
    cfg = dict(
        model_type="qwen3" if qwen else
        ("mistral" if rng.random() < 0.3 else "llama"),
        hidden_size=int(rng.choice([64, 128, 192, 256])),
        intermediate_size=int(rng.choice([64, 128, 256, 384])),
        vocab_size=int(rng.choice([97, 256, 513, 1000])),
        num_hidden_layers=layers,
        num_attention_heads=nh, num_key_value_heads=nkv, head_dim=hd,
        rms_norm_eps=float(rng.choice([1e-5, 1e-6])),
        rope_theta=float(rng.choice([10000.0, 500000.0, 1000000.0])),
        max_position_embeddings=512,
        tie_word_embeddings=bool(rng.random() < 0.4),
    )
    if rng.random() < 0.35:
        cfg.update(sliding_window=int(rng.choice([8, 24, 48])),
                   use_sliding_window=True,
                   max_window_layers=int(rng.integers(0, layers + 1)))
    if cfg["model_type"] == "llama" and rng.random() < 0.3:
        cfg["rope_scaling"] = dict(
            rope_type="llama3", factor=8.0, low_freq_factor=1.0,
            high_freq_factor=4.0, original_max_position_embeddings=64)
    return cfg


def run_one(i, cfg_json, rng):
    cfg = Config.from_json(cfg_json)
    w = random_weights(cfg, seed=1000 + i)
    oracle = quantized_oracle(cfg, w)
    plen = int(rng.integers(3, 90))
    prompt = rng.integers(0, cfg.vocab_size, size=plen).astype(np.uint32)
    with tempfile.TemporaryDirectory() as td:
        st = os.path.join(td, "m.safetensors")
        weights_to_safetensors(w, cfg, st)
        # max_batch_tokens 32 forces CHUNKED prefill for most prompts
        eng = cake_amd.Engine(json.dumps(cfg_json), max_seq=256,
                              max_batch_tokens=32)
        eng.load_safetensors(st)
        try:
            first, lg = eng.prefill(prompt, want_logits=True)
            ref = oracle.forward(prompt[None, :].astype(np.int64), 0)[0]
            r1 = rel_err(lg, ref)
            toks = eng.decode(5)
            seq = np.concatenate([prompt, [first], toks[:-1]]).astype(
                np.uint32)
            eng.reset()
            _, lg2 = eng.prefill(seq, want_logits=True)
            consistent = int(np.argmax(lg2)) == int(toks[-1])
            oracle.reset()
            ref2 = oracle.forward(seq[None, :].astype(np.int64), 0)[0]
            r2 = rel_err(lg2, ref2)
            ok = r1 < 2e-2 and r2 < 2e-2 and consistent
            tag = "ok " if ok else "FAIL"
            print(f"[{i:02d}] {tag} rel={r1:.1e}/{r2:.1e} "
                  f"dec-consistent={consistent} "
                  f"{cfg_json['model_type']} nh={cfg.num_attention_heads}"
                  f"/{cfg.num_key_value_heads} hd={cfg.hd} "
                  f"L={cfg.num_hidden_layers} "
                  f"win={cfg_json.get('sliding_window')}"
                  f"@mwl{cfg_json.get('max_window_layers')} "
                  f"tied={cfg_json['tie_word_embeddings']} plen={plen}",
                  flush=True)
            return ok
        finally:
            eng.close()


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 24
    rng = np.random.default_rng(299792458)
    fails = 0
    for i in range(n):
        cfg = random_cfg(rng)
        try:
            if not run_one(i, cfg, rng):
                fails += 1
        except Exception as e:
            fails += 1
            print(f"[{i:02d}] EXC {type(e).__name__}: {e} — cfg={cfg}",
                  flush=True)
    print(f"FUZZ {'FAILED' if fails else 'PASSED'}: {n} configs, "
          f"{fails} failures")
    sys.exit(1 if fails else 0)


if __name__ == "__main__":
    main()
