"""BASELINE.json model configurations (architecture shapes only; weights are
random-init — no network for checkpoints).  Shapes per SURVEY.md §8 header."""

LLAMA3_8B = dict(
    model_type="llama", hidden_size=4096, intermediate_size=14336,
    vocab_size=128256, num_hidden_layers=32, num_attention_heads=32,
    num_key_value_heads=8, head_dim=128, rms_norm_eps=1e-5,
    rope_theta=500000.0, max_position_embeddings=8192,
    tie_word_embeddings=False,
    rope_scaling=dict(rope_type="llama3", factor=8.0, low_freq_factor=1.0,
                      high_freq_factor=4.0,
                      original_max_position_embeddings=8192))

LLAMA3_70B = dict(
    model_type="llama", hidden_size=8192, intermediate_size=28672,
    vocab_size=128256, num_hidden_layers=80, num_attention_heads=64,
    num_key_value_heads=8, head_dim=128, rms_norm_eps=1e-5,
    rope_theta=500000.0, max_position_embeddings=8192,
    tie_word_embeddings=False,
    rope_scaling=dict(rope_type="llama3", factor=8.0, low_freq_factor=1.0,
                      high_freq_factor=4.0,
                      original_max_position_embeddings=8192))

QWEN3_0_6B = dict(
    model_type="qwen3", hidden_size=1024, intermediate_size=3072,
    vocab_size=151936, num_hidden_layers=28, num_attention_heads=16,
    num_key_value_heads=8, head_dim=128, rms_norm_eps=1e-6,
    rope_theta=1000000.0, max_position_embeddings=8192,
    tie_word_embeddings=True)

QWEN3_32B = dict(
    model_type="qwen3", hidden_size=5120, intermediate_size=25600,
    vocab_size=151936, num_hidden_layers=64, num_attention_heads=64,
    num_key_value_heads=8, head_dim=128, rms_norm_eps=1e-6,
    rope_theta=1000000.0, max_position_embeddings=8192,
    tie_word_embeddings=False)

QWEN3_32B_FP8 = dict(QWEN3_32B,
    quantization_config=dict(quant_method="fp8",
                             weight_block_size=[128, 128]))

# Mistral-7B-v0.1 shape (mistral/, 196 LoC of the same common Transformer +
# the sliding-window flag cache.rs:173-205 consumes; window now runs in the
# attention kernels' span bound)
MISTRAL_7B = dict(
    model_type="mistral", hidden_size=4096, intermediate_size=14336,
    vocab_size=32000, num_hidden_layers=32, num_attention_heads=32,
    num_key_value_heads=8, head_dim=128, rms_norm_eps=1e-5,
    rope_theta=10000.0, max_position_embeddings=32768,
    tie_word_embeddings=False, sliding_window=4096,
)

MODELS = {
    "llama3-8b": LLAMA3_8B,
    "llama3-70b": LLAMA3_70B,
    "qwen3-0.6b": QWEN3_0_6B,
    "mistral-7b": MISTRAL_7B,
    "qwen3-32b": QWEN3_32B,
    "qwen3-32b-fp8": QWEN3_32B_FP8,
}


def weight_bytes_bf16(cfg: dict, lo=0, hi=None, embed=True, head=True) -> int:
    H, I, V = cfg["hidden_size"], cfg["intermediate_size"], cfg["vocab_size"]
    nh, nkv = cfg["num_attention_heads"], cfg["num_key_value_heads"]
    hd = cfg.get("head_dim") or H // nh
    L = (hi if hi is not None else cfg["num_hidden_layers"]) - lo
    per_layer = (nh + 2 * nkv) * hd * H + H * nh * hd + 3 * I * H + 2 * H
    total = L * per_layer
    if embed:
        total += V * H
    if head:
        total += H
        if not cfg.get("tie_word_embeddings"):
            total += V * H
    return total * 2
