"""CPU oracle — restatement of cake's hot-path arithmetic (TEST INFRASTRUCTURE ONLY).

This package is the parity oracle for the MI355X-native engine in `cake_amd/`.
It restates, in numpy f32, exactly the math the reference (evilsocket/cake @
/root/reference) performs on its layer-sharded LLM decode/prefill hot path.

ONLY `tests/`, `__graft_entry__.smoke()` and `bench.py`'s cpu_baseline leg may
import or call this package — never the product path.  The product path is the
HIP engine behind `include/cake_hip.h`; it fails loudly if its extension is
missing, and never falls back to this code.

Reference citations (file:line are into /root/reference/cake-core/src):
  - rms_norm:            backends/mod.rs:244-246 (candle_nn::ops::rms_norm:
                         x * w / sqrt(mean(x^2) + eps), f32 accumulate)
  - linear_forward:      backends/mod.rs:206-241 (x @ W^T, optional bias)
  - softmax (last dim):  backends/mod.rs:407-440 (max-subtract, exp, sum, div; f32)
  - rope:                backends/mod.rs:444-482 (HF half-rotation:
                         out[i] = x1*c - x2*s; out[i+half] = x2*c + x1*s)
  - cos/sin tables:      models/common/cache.rs:31-99 (theta_i =
                         rope_theta^(-2i/rotary_dim); llama3 scaling 49-80;
                         idx_theta = pos * theta; NOTE cake casts tables to the
                         model dtype (cache.rs:95-96) — the oracle keeps f32,
                         matching HF transformers; difference is inside the
                         stated bf16 tolerance)
  - attention:           models/common/attention.rs:152-357 (fused QKV narrow-
                         split 166-174, optional per-head QK-norm 202-215,
                         reshape/transpose 194-239, RoPE 242-253, KV append
                         255-262, f32 GQA attention: repeat_kv + QK^T/sqrt(d) +
                         causal mask + softmax + PV at 300-343, o_proj 354)
  - causal mask:         models/common/cache.rs:150-160 (u8, 1 where j > i) and
                         attention.rs:324-333 (zero-pad left when kv_len > seq)
  - KV cache append:     models/common/cache.rs:184-210 (cat on dim 2)
  - MLP (SwiGLU):        models/common/mlp.rs:21-31 (fused gate_up, narrow,
                         silu(gate)*up, down_proj)
  - silu_mul:            backends/mod.rs:82 + backends/cuda/ops.cu:101-138
                         (x * sigmoid(x) * y)
  - transformer block:   models/common/transformer.rs:103-135 (pre-norm:
                         h = x + attn(rms1(x)); out = h + mlp(rms2(h)))
  - full forward:        models/common/text_model.rs:266-368 (embedding ->
                         block loop -> final rms_norm -> last-token slice ->
                         lm_head)
  - greedy sampling:     models/common/text_model.rs:102-118 (ArgMax when
                         temperature <= 0)

Parity pinning: the reference itself cannot be compiled here (Rust toolchain
absent) and its numeric ground truth lives in the un-vendored candle 0.9.2
(Cargo.lock), which tracks HF transformers semantics (cache.rs:93-94 cites HF
modeling_llama.py).  This oracle is therefore cross-validated against HF
transformers 5.15 on torch-2.10 CPU (see oracle/gen_golden.py), and the
resulting golden vectors are committed under tests/golden/.
"""

from dataclasses import dataclass, field
from typing import Optional

import numpy as np


# ---------------------------------------------------------------------------
# Config — mirrors models/common/config.rs:87-153 (hot-path subset) and the
# config.json auto-detection in cake/mod.rs:82-110,268-274.
# ---------------------------------------------------------------------------
@dataclass
class Config:
    hidden_size: int
    intermediate_size: int
    vocab_size: int
    num_hidden_layers: int
    num_attention_heads: int
    num_key_value_heads: int
    rms_norm_eps: float = 1e-5
    rope_theta: float = 500000.0
    max_seq_len: int = 4096
    head_dim: Optional[int] = None          # config.rs:106
    tie_word_embeddings: bool = False       # config.rs:99
    use_qk_norm: bool = False               # config.rs:115 (Qwen3: true)
    partial_rotary_factor: float = 1.0      # config.rs:108
    # llama3 rope scaling (config.rs:50-65); None = no scaling
    rope_scaling: Optional[dict] = None
    model_prefix: str = "model"             # config.rs:104
    # sliding-window attention span (cache.rs:173-205 trims KV to the
    # window; bounding the attention span is equivalent); None = full.
    # layer_windows: per-layer span (Gemma3/EXAONE-style interleave of
    # local and global layers; qwen-style max_window_layers) — tuple of
    # (window-or-None) per layer, built by from_json
    sliding_window: Optional[int] = None
    layer_windows: Optional[tuple] = None

    def window_for(self, block_idx: int) -> Optional[int]:
        if self.layer_windows is not None:
            return self.layer_windows[block_idx]
        return self.sliding_window

    @property
    def hd(self) -> int:
        # attention.rs:85
        return self.head_dim if self.head_dim is not None else (
            self.hidden_size // self.num_attention_heads)

    @property
    def rotary_dim(self) -> int:
        # attention.rs:86
        return int(self.hd * self.partial_rotary_factor)

    @classmethod
    def from_json(cls, j: dict) -> "Config":
        """Build from an HF-style config.json dict (cake/mod.rs:268-274)."""
        rs = j.get("rope_scaling")
        if rs is not None and rs.get("rope_type", rs.get("type")) != "llama3":
            rs = None
        return cls(
            hidden_size=j["hidden_size"],
            intermediate_size=j["intermediate_size"],
            vocab_size=j["vocab_size"],
            num_hidden_layers=j["num_hidden_layers"],
            num_attention_heads=j["num_attention_heads"],
            num_key_value_heads=j.get("num_key_value_heads",
                                      j["num_attention_heads"]),
            rms_norm_eps=j.get("rms_norm_eps", 1e-5),
            rope_theta=j.get("rope_theta", 10000.0),
            max_seq_len=j.get("max_position_embeddings", 4096),
            head_dim=j.get("head_dim"),
            tie_word_embeddings=j.get("tie_word_embeddings", False),
            use_qk_norm="qwen3" in j.get("model_type", ""),
            rope_scaling=rs,
            sliding_window=(j.get("sliding_window")
                            if j.get("use_sliding_window", True) else None),
            layer_windows=cls._layer_windows(j),
        )

    @staticmethod
    def _layer_windows(j: dict) -> Optional[tuple]:
        '''Per-layer attention window.  HF "layer_types" wins
        ("sliding_attention"/"full_attention" per layer); otherwise the
        qwen-style rule: sliding for layer_idx >= max_window_layers when
        use_sliding_window, all layers when only sliding_window is set
        (mistral).'''
        w = j.get("sliding_window")
        if not w or not j.get("use_sliding_window", True):
            return None
        L = j["num_hidden_layers"]
        lt = j.get("layer_types")
        if lt is not None:
            return tuple(w if t == "sliding_attention" else None
                         for t in lt)
        mwl = j.get("max_window_layers", 0)
        return tuple(w if i >= mwl else None for i in range(L))


# ---------------------------------------------------------------------------
# Primitive ops
# ---------------------------------------------------------------------------
def rms_norm(x: np.ndarray, w: np.ndarray, eps: float) -> np.ndarray:
    """backends/mod.rs:244-246: x * w / sqrt(mean(x^2, last dim) + eps)."""
    x = x.astype(np.float32)
    ms = np.mean(x * x, axis=-1, keepdims=True)
    return (x / np.sqrt(ms + eps)) * w.astype(np.float32)


def silu_mul(gate: np.ndarray, up: np.ndarray) -> np.ndarray:
    """backends/mod.rs:82 / ops.cu:101-138: gate*sigmoid(gate)*up."""
    g = gate.astype(np.float32)
    return (g / (1.0 + np.exp(-g))) * up.astype(np.float32)


def softmax_lastdim(x: np.ndarray) -> np.ndarray:
    """backends/mod.rs:407-440 fast path (f32, last dim)."""
    m = np.max(x, axis=-1, keepdims=True)
    e = np.exp(x - m)
    return e / np.sum(e, axis=-1, keepdims=True)


def linear(x: np.ndarray, w: np.ndarray, b: Optional[np.ndarray] = None
           ) -> np.ndarray:
    """backends/mod.rs:206-241: x @ W^T (+ bias). W is (out, in)."""
    y = x.astype(np.float32) @ w.astype(np.float32).T
    if b is not None:
        y = y + b.astype(np.float32)
    return y


def rope_tables(cfg: Config) -> tuple[np.ndarray, np.ndarray]:
    """cache.rs:43-99 — (max_seq_len, rotary_dim/2) f32 cos/sin tables.

    theta_i = 1 / rope_theta^(i/rotary_dim) for i = 0,2,4,... (cache.rs:43-46)
    with optional llama3 frequency scaling (cache.rs:49-80).
    """
    rd = cfg.rotary_dim
    i = np.arange(0, rd, 2, dtype=np.float32)
    theta = (1.0 / np.power(np.float32(cfg.rope_theta), i / np.float32(rd))
             ).astype(np.float32)
    if cfg.rope_scaling is not None:
        # cache.rs:54-79
        factor = np.float32(cfg.rope_scaling["factor"])
        lo_f = np.float32(cfg.rope_scaling["low_freq_factor"])
        hi_f = np.float32(cfg.rope_scaling["high_freq_factor"])
        old_ctx = np.float32(
            cfg.rope_scaling["original_max_position_embeddings"])
        low_wl = old_ctx / lo_f
        high_wl = old_ctx / hi_f
        out = theta.copy()
        for n, f in enumerate(theta):
            wavelen = 2.0 * np.float32(np.pi) / f
            if wavelen < high_wl:
                pass                      # high frequency: keep
            elif wavelen > low_wl:
                out[n] = f / factor       # low frequency: scale down
            else:
                smooth = (old_ctx / wavelen - lo_f) / (hi_f - lo_f)
                out[n] = (1.0 - smooth) * (f / factor) + smooth * f
        theta = out
    pos = np.arange(cfg.max_seq_len, dtype=np.float32)[:, None]
    idx_theta = pos * theta[None, :]      # cache.rs:86-89
    return np.cos(idx_theta), np.sin(idx_theta)


def rope(x: np.ndarray, cos: np.ndarray, sin: np.ndarray) -> np.ndarray:
    """backends/mod.rs:444-482 — HF half-rotation.

    x: (B, H, S, D); cos/sin: (S, D/2) for the S positions of x.
    out[..., i]      = x[..., i]   * c[i] - x[..., i+h] * s[i]
    out[..., i + h]  = x[..., i+h] * c[i] + x[..., i]   * s[i]
    """
    half = x.shape[-1] // 2
    x1 = x[..., :half]
    x2 = x[..., half:]
    c = cos[None, None, :, :]
    s = sin[None, None, :, :]
    return np.concatenate([x1 * c - x2 * s, x2 * c + x1 * s], axis=-1)


def causal_mask(seq_len: int, kv_len: int,
                window: Optional[int] = None) -> np.ndarray:
    """cache.rs:150-160 + attention.rs:324-333: True where masked.

    Causal (j > pos_i) over absolute positions pos_i = kv_len - seq + i,
    left-padded with zeros (attend) when kv_len > seq_len (prefill into an
    existing cache).  With a sliding window W, positions j <= pos_i - W are
    masked too (cache.rs:173-205 trims the KV to the window; masking the
    span is equivalent)."""
    pos = kv_len - seq_len + np.arange(seq_len)[:, None]
    j = np.arange(kv_len)[None, :]
    m = j > pos
    if window:
        m = m | (j <= pos - window)
    return m


# ---------------------------------------------------------------------------
# Weights containers
# ---------------------------------------------------------------------------
@dataclass
class LayerWeights:
    """One transformer block's tensors, HF naming (transformer.rs:79-101)."""
    input_layernorm: np.ndarray          # (H,)
    post_attention_layernorm: np.ndarray  # (H,)
    q_proj: np.ndarray                   # (n_heads*hd, H)
    k_proj: np.ndarray                   # (n_kv*hd, H)
    v_proj: np.ndarray                   # (n_kv*hd, H)
    o_proj: np.ndarray                   # (H, n_heads*hd)
    gate_proj: np.ndarray                # (I, H)
    up_proj: np.ndarray                  # (I, H)
    down_proj: np.ndarray                # (H, I)
    q_norm: Optional[np.ndarray] = None  # (hd,) — Qwen3 (attention.rs:120-129)
    k_norm: Optional[np.ndarray] = None  # (hd,)


@dataclass
class ModelWeights:
    embed_tokens: np.ndarray             # (V, H)
    norm: np.ndarray                     # (H,)
    lm_head: np.ndarray                  # (V, H); == embed_tokens when tied
    layers: list = field(default_factory=list)


# ---------------------------------------------------------------------------
# The model — mirrors TextModelBase + Transformer + CausalSelfAttention + MLP
# ---------------------------------------------------------------------------
class OracleModel:
    """Greedy decode/prefill oracle following cake's exact op order."""

    def __init__(self, cfg: Config, w: ModelWeights):
        self.cfg = cfg
        self.w = w
        self.cos, self.sin = rope_tables(cfg)
        # KV cache: per layer list of (K, V) with shape (1, n_kv, n, hd)
        # (cache.rs:15, 184-210)
        self.kv: list = [None] * cfg.num_hidden_layers

    def reset(self):
        self.kv = [None] * self.cfg.num_hidden_layers

    # -- attention.rs:152-357 -------------------------------------------------
    def attention(self, lw: LayerWeights, x: np.ndarray, index_pos: int,
                  block_idx: int) -> np.ndarray:
        cfg = self.cfg
        b, s, _ = x.shape
        hd, nh, nkv = cfg.hd, cfg.num_attention_heads, cfg.num_key_value_heads
        size_q, size_kv = nh * hd, nkv * hd
        # fused QKV (attention.rs:109-114,162-174)
        qkv_w = np.concatenate([lw.q_proj, lw.k_proj, lw.v_proj], axis=0)
        qkv = linear(x, qkv_w)
        q = qkv[..., :size_q]
        k = qkv[..., size_q:size_q + size_kv]
        v = qkv[..., size_q + size_kv:]
        # reshape (b, s, h, hd) (attention.rs:194-199)
        q = q.reshape(b, s, nh, hd)
        k = k.reshape(b, s, nkv, hd)
        v = v.reshape(b, s, nkv, hd)
        # per-head QK-norm on head_dim (attention.rs:202-215)
        if lw.q_norm is not None:
            q = rms_norm(q, lw.q_norm, cfg.rms_norm_eps)
        if lw.k_norm is not None:
            k = rms_norm(k, lw.k_norm, cfg.rms_norm_eps)
        # transpose to (b, h, s, hd) (attention.rs:221-239)
        q = q.transpose(0, 2, 1, 3)
        k = k.transpose(0, 2, 1, 3)
        v = v.transpose(0, 2, 1, 3)
        # RoPE with cos/sin rows [index_pos, index_pos+s) (attention.rs:50-68)
        c = self.cos[index_pos:index_pos + s]
        sn = self.sin[index_pos:index_pos + s]
        q = rope(q, c, sn)
        k = rope(k, c, sn)
        # KV append (cache.rs:184-210)
        if self.kv[block_idx] is not None:
            ck, cv = self.kv[block_idx]
            k = np.concatenate([ck, k], axis=2)
            v = np.concatenate([cv, v], axis=2)
        self.kv[block_idx] = (k, v)
        kv_len = k.shape[2]
        # GQA repeat (attention.rs:305-311)
        rep = nh // nkv
        kr = np.repeat(k, rep, axis=1)
        vr = np.repeat(v, rep, axis=1)
        # f32 attention (attention.rs:300-343)
        att = q @ kr.transpose(0, 1, 3, 2) / np.float32(np.sqrt(hd))
        win = self.cfg.window_for(block_idx)
        if s > 1 or win:
            m = causal_mask(s, kv_len, win)
            att = np.where(m[None, None], np.float32(-np.inf), att)
        att = softmax_lastdim(att)
        y = att @ vr
        # back to (b, s, size_q) (attention.rs:349-353) then o_proj (354)
        y = y.transpose(0, 2, 1, 3).reshape(b, s, size_q)
        return linear(y, lw.o_proj)

    # -- mlp.rs:21-31 ---------------------------------------------------------
    def mlp(self, lw: LayerWeights, x: np.ndarray) -> np.ndarray:
        gate = linear(x, lw.gate_proj)
        up = linear(x, lw.up_proj)
        return linear(silu_mul(gate, up), lw.down_proj)

    # -- transformer.rs:103-135 ----------------------------------------------
    def block(self, i: int, x: np.ndarray, index_pos: int) -> np.ndarray:
        lw = self.w.layers[i]
        h = rms_norm(x, lw.input_layernorm, self.cfg.rms_norm_eps)
        x = x + self.attention(lw, h, index_pos, i)
        h = rms_norm(x, lw.post_attention_layernorm, self.cfg.rms_norm_eps)
        return x + self.mlp(lw, h)

    # -- text_model.rs:266-368 ------------------------------------------------
    def forward(self, token_ids: np.ndarray, index_pos: int) -> np.ndarray:
        """token_ids: (1, S) -> logits (1, V) for the LAST position, f32."""
        x = self.w.embed_tokens[token_ids].astype(np.float32)  # embedding
        for i in range(self.cfg.num_hidden_layers):
            x = self.block(i, x, index_pos)
        x = rms_norm(x, self.w.norm, self.cfg.rms_norm_eps)
        x = x[:, -1, :]                                    # last-token slice
        return linear(x, self.w.lm_head)

    def hidden_forward(self, x: np.ndarray, index_pos: int,
                       lo: int, hi: int) -> np.ndarray:
        """Run blocks [lo, hi) on hidden states — the Forwarder::forward_batch
        unit (cake/mod.rs:533-540, worker.rs:299-578)."""
        for i in range(lo, hi):
            x = self.block(i, x, index_pos)
        return x

    # -- text_model.rs:397-495 (greedy) ---------------------------------------
    def generate_greedy(self, prompt_ids: list, max_new: int) -> list:
        self.reset()
        tokens = list(prompt_ids)
        index_pos = 0
        out = []
        for step in range(max_new):
            if step == 0:
                ctx = tokens
            else:
                ctx = tokens[-1:]
            logits = self.forward(np.array([ctx], dtype=np.int64), index_pos)
            index_pos += len(ctx)
            nxt = int(np.argmax(logits[0]))    # ArgMax (text_model.rs:104)
            tokens.append(nxt)
            out.append(nxt)
        return out


# ---------------------------------------------------------------------------
# Deterministic random weights (shared by tests and the GPU engine's
# safetensors files — see tests/helpers.py)
# ---------------------------------------------------------------------------
def random_weights(cfg: Config, seed: int = 299792458,
                   scale: float = 0.02) -> ModelWeights:
    """Seeded-random model weights; seed default = cake's (lib.rs:180)."""
    rng = np.random.default_rng(seed)
    H, I, V = cfg.hidden_size, cfg.intermediate_size, cfg.vocab_size
    hd, nh, nkv = cfg.hd, cfg.num_attention_heads, cfg.num_key_value_heads

    def t(*shape):
        return (rng.standard_normal(shape) * scale).astype(np.float32)

    layers = []
    for _ in range(cfg.num_hidden_layers):
        layers.append(LayerWeights(
            input_layernorm=1.0 + t(H),
            post_attention_layernorm=1.0 + t(H),
            q_proj=t(nh * hd, H), k_proj=t(nkv * hd, H), v_proj=t(nkv * hd, H),
            o_proj=t(H, nh * hd),
            gate_proj=t(I, H), up_proj=t(I, H), down_proj=t(H, I),
            q_norm=(1.0 + t(hd)) if cfg.use_qk_norm else None,
            k_norm=(1.0 + t(hd)) if cfg.use_qk_norm else None,
        ))
    embed = t(V, H)
    return ModelWeights(
        embed_tokens=embed,
        norm=1.0 + t(H),
        lm_head=embed if cfg.tie_word_embeddings else t(V, H),
        layers=layers,
    )


# ---------------------------------------------------------------------------
# FP8 (float8_e4m3fn) blockwise dequantization — restates utils/fp8.rs:42-64:
#   w[i, j] = f8_to_f32(w8[i, j]) * scale_inv[i // 128, j // 128]
# E4M3FN: 1 sign, 4 exp (bias 7), 3 mantissa; NaN = 0x7F/0xFF; no inf.
# (gfx950 uses the same OCP e4m3fn encoding.)
# ---------------------------------------------------------------------------
FP8_BLOCK = 128  # fp8.rs:17


def f8e4m3_to_f32(b: np.ndarray) -> np.ndarray:
    """Decode uint8 e4m3fn -> f32 (vectorized restatement of the bit
    decode in backends/cuda/ops.cu:31-51)."""
    b = b.astype(np.uint32)
    sign = np.where(b & 0x80, np.float32(-1.0), np.float32(1.0))
    e = (b >> 3) & 0xF
    m = b & 0x7
    nan = (e == 15) & (m == 7)
    val = np.where(
        e == 0,
        m.astype(np.float32) * np.float32(2.0 ** -9),          # subnormal
        (1.0 + m.astype(np.float32) / 8.0) *
        np.exp2((e.astype(np.float32) - 7.0)))
    out = (sign * val).astype(np.float32)
    out[nan] = np.nan
    return out


def fp8_dequant(w8: np.ndarray, scale_inv: np.ndarray) -> np.ndarray:
    """utils/fp8.rs:42-64 — blockwise 128x128 dequant to f32."""
    n, k = w8.shape
    f = f8e4m3_to_f32(w8)
    bi = np.arange(n) // FP8_BLOCK
    bj = np.arange(k) // FP8_BLOCK
    return (f * scale_inv.astype(np.float32)[np.ix_(bi, bj)]).astype(
        np.float32)
