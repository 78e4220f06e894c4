import json, sys, os
sys.path.insert(0, '.')
import numpy as np
import cake_amd
from oracle import Config, random_weights, OracleModel
from tests.helpers import weights_to_safetensors
import tempfile

cfg_json = dict(model_type="llama", hidden_size=256, intermediate_size=512,
    vocab_size=512, num_hidden_layers=2, num_attention_heads=4,
    num_key_value_heads=2, head_dim=64, rms_norm_eps=1e-5,
    rope_theta=10000.0, max_position_embeddings=128, tie_word_embeddings=False)
cfg = Config.from_json(cfg_json)
w = random_weights(cfg, seed=7)
oracle = OracleModel(cfg, w)
rng = np.random.default_rng(42)
prompt = rng.integers(0, cfg.vocab_size, size=9).astype(np.int64)
ref_tokens = oracle.generate_greedy(list(prompt), 6)
with tempfile.TemporaryDirectory() as td:
    st = os.path.join(td, "model.safetensors")
    weights_to_safetensors(w, cfg, st)
    flags = cake_amd.HAS_EMBED | cake_amd.HAS_HEAD
    if os.environ.get("PROBE_GRAPH", "1") == "1":
        flags |= cake_amd.USE_GRAPH
    eng = cake_amd.Engine(json.dumps(cfg_json), max_seq=64,
                          max_batch_tokens=32, flags=flags)
    eng.load_safetensors(st)
    first, _ = eng.prefill(prompt.astype(np.uint32), want_logits=True)
    print("prefill ok, first =", first, flush=True)
    toks = [int(first)] + [int(t) for t in eng.decode(5)]
    print("gpu   ", toks, flush=True)
    print("oracle", ref_tokens, flush=True)
    eng.close()
print("PROBE OK")
