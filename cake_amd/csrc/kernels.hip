// cake_hip kernels — hand-written HIP for gfx950 (MI355X, CDNA4).
//
// These implement the compute ops of cake's hot path:
//   rms_norm        backends/mod.rs:244-246
//   linear_forward  backends/mod.rs:206-241  (GEMV for decode M=1, MFMA GEMM
//                                             for prefill)
//   rope            backends/mod.rs:444-482  (HF half-rotation)
//   silu_mul        backends/mod.rs:82, backends/cuda/ops.cu:101-138
//   attention       models/common/attention.rs:266-343 (f32 softmax, GQA)
//   KV cache        models/common/cache.rs:184-210 — REDESIGNED as a
//                   preallocated device cache (fixes the O(n^2) cat-per-token)
//   embedding       backends/mod.rs:513-528
//   argmax          models/common/text_model.rs:102-118 (greedy ArgMax)
//
// Design notes (MI355X-first, per /opt/skills/guides/cdna_hip_programming.md):
//   - wave = 64 lanes everywhere; blocks are multiples of 64
//   - bf16 storage, f32 accumulation (matches cake's CUDA dtype policy)
//   - decode GEMV: HBM-bound; 16 B/lane coalesced weight streaming, weights
//     loaded straight to VGPRs (guide §5 "GEMV / M <= 16": no LDS round trip)
//   - prefill GEMM: MFMA v_mfma_f32_16x16x32_bf16, 128x128 tile, 4 waves,
//     double-buffered LDS filled by global_load_lds (16 B) with the XOR
//     source-swizzle (guide §5.4 rule 21) so ds_read_b128 is conflict-free
//   - decode attention: split-KV online softmax, two-pass combine; K/V reads
//     are 64-lane x 4 B = one 256 B transaction per row (coalesced)
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>
#include <cstdlib>

#define WAVE 64

using u16 = unsigned short;
using u32 = unsigned int;
using f32x4 = __attribute__((ext_vector_type(4))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using short8 = __attribute__((ext_vector_type(8))) short;

// ---------------------------------------------------------------------------
// bf16 <-> f32 (round-to-nearest-even, matches torch/HF casting)
// ---------------------------------------------------------------------------
__device__ __host__ inline u16 f2b(float f) {
  union { float f; u32 u; } v{f};
  if ((v.u & 0x7fffffffu) > 0x7f800000u) return 0x7fc0;  // NaN
  u32 r = v.u + 0x7fffu + ((v.u >> 16) & 1u);
  return (u16)(r >> 16);
}
__device__ __host__ inline float b2f(u16 h) {
  union { u32 u; float f; } v{(u32)h << 16};
  return v.f;
}

// nontemporal 16-B weight load: decode weights are streamed exactly once per
// token — keep them from thrashing L2/L3 (guide G14: streaming data)
using uint4v = __attribute__((ext_vector_type(4))) unsigned int;
__device__ inline short8 ntload8(const u16* p) {
  uint4v v = __builtin_nontemporal_load(reinterpret_cast<const uint4v*>(p));
  union { uint4v u; short8 s; } c{v};
  return c.s;
}

// ---------------------------------------------------------------------------
// FP8 e4m3fn decode (OCP encoding, gfx950-native format; restates the bit
// decode of backends/cuda/ops.cu:31-51 / utils/fp8.rs): place sign at bit
// 31 and exp|man at bits 20-26, then scale by 2^120 — handles normals AND
// subnormals in two ops (f32 denormals are not flushed in the default
// kernel mode, guide §3).  The NaN encodings (0x7f/0xff) decode to a large
// finite value; real checkpoints contain no NaN weights (documented).
// ---------------------------------------------------------------------------
__device__ inline float f8tof(unsigned char b) {
  union { u32 u; float f; } v;
  v.u = ((u32)(b & 0x80u) << 24) | ((u32)(b & 0x7fu) << 20);
  return v.f * 0x1p120f;
}
using f32x2 = __attribute__((ext_vector_type(2))) float;
__device__ inline uint4v ntload16b(const unsigned char* p) {
  return __builtin_nontemporal_load(reinterpret_cast<const uint4v*>(p));
}
// 16 fp8 -> 16 f32 via the hardware v_cvt_pk_f32_fp8 (2 elems/instruction;
// a bit-twiddle decode is VALU-bound at ~1 TB/s)
__device__ inline void f8x16_decode(uint4v w, float* out) {
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    f32x2 lo = __builtin_amdgcn_cvt_pk_f32_fp8(w[q], false);
    f32x2 hi = __builtin_amdgcn_cvt_pk_f32_fp8(w[q], true);
    out[q * 4 + 0] = lo[0];
    out[q * 4 + 1] = lo[1];
    out[q * 4 + 2] = hi[0];
    out[q * 4 + 3] = hi[1];
  }
}

__device__ inline float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, WAVE);
  return v;
}
__device__ inline float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_down(v, off, WAVE));
  return v;
}

// ---------------------------------------------------------------------------
// conversion / fill
// ---------------------------------------------------------------------------
__global__ void k_f32_to_bf16(const float* __restrict__ in,
                              u16* __restrict__ out, size_t n) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) out[i] = f2b(in[i]);
}
__global__ void k_bf16_to_f32(const u16* __restrict__ in,
                              float* __restrict__ out, size_t n) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) out[i] = b2f(in[i]);
}

// splitmix64-based uniform(-1,1)*scale fill (deterministic per element)
__global__ void k_fill_random(u16* __restrict__ out, size_t n, uint64_t seed,
                              float scale) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    uint64_t z = seed + 0x9e3779b97f4a7c15ull * (i + 1);
    z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ull;
    z = (z ^ (z >> 27)) * 0x94d049bb133111ebull;
    z ^= z >> 31;
    float u = (float)(z >> 40) * (1.0f / 8388608.0f) - 1.0f;  // [-1, 1)
    out[i] = f2b(u * scale);
  }
}
// random e4m3fn bytes for synthetic fp8 weights: NaN encodings (0x7f/0xff)
// are remapped into the largest normal
__global__ void k_fill_random_u8(unsigned char* __restrict__ out, size_t n,
                                 uint64_t seed) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    uint64_t z = seed + 0x9e3779b97f4a7c15ull * (i + 1);
    z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ull;
    z = (z ^ (z >> 27)) * 0x94d049bb133111ebull;
    z ^= z >> 31;
    unsigned char b = (unsigned char)(z >> 56);
    if ((b & 0x7f) == 0x7f) b ^= 0x08;  // avoid NaN encodings
    out[i] = b;
  }
}
__global__ void k_fill_const(u16* __restrict__ out, size_t n, float v) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  u16 b = f2b(v);
  for (; i < n; i += stride) out[i] = b;
}

// ---------------------------------------------------------------------------
// rms_norm (backends/mod.rs:244-246): out = x * w / sqrt(mean(x^2) + eps)
// one block per row; bf16x8 vector loads (guide G13); f32 accumulate
// ---------------------------------------------------------------------------
// rows are addressed as (outer, inner): row r -> offset
// (r/inner)*outer_stride + (r%inner)*cols.  Plain contiguous rows use
// inner = nrows, outer_stride = 0.  The (outer, inner) form handles the
// per-head QK-norm rows inside the strided qkv buffer
// (attention.rs:202-215: norm over head_dim after reshape).
__global__ void k_rmsnorm(const u16* __restrict__ x, const u16* __restrict__ w,
                          u16* __restrict__ out, int cols, float eps,
                          int inner, size_t outer_stride) {
  const int row = blockIdx.x;
  const size_t off =
      (size_t)(row / inner) * outer_stride + (size_t)(row % inner) * cols;
  const u16* xr = x + off;
  u16* outr = out + off;
  float ss = 0.f;
  const int t = threadIdx.x;
  const int nvec = cols / 8;
  for (int i = t; i < nvec; i += blockDim.x) {
    short8 v = *reinterpret_cast<const short8*>(xr + i * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = b2f((u16)v[j]);
      ss += f * f;
    }
  }
  for (int i = nvec * 8 + t; i < cols; i += blockDim.x) {
    float f = b2f(xr[i]);
    ss += f * f;
  }
  ss = wave_sum(ss);
  __shared__ float red[16];
  const int wid = t / WAVE, lane = t % WAVE;
  if (lane == 0) red[wid] = ss;
  __syncthreads();
  const int nw = blockDim.x / WAVE;
  float tot = 0.f;
#pragma unroll
  for (int i = 0; i < 16; ++i)
    if (i < nw) tot += red[i];
  const float scale = rsqrtf(tot / (float)cols + eps);
  for (int i = t; i < nvec; i += blockDim.x) {
    short8 v = *reinterpret_cast<const short8*>(xr + i * 8);
    short8 wv = *reinterpret_cast<const short8*>(w + i * 8);
    short8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = (short)f2b(b2f((u16)v[j]) * scale * b2f((u16)wv[j]));
    *reinterpret_cast<short8*>(outr + i * 8) = o;
  }
  for (int i = nvec * 8 + t; i < cols; i += blockDim.x)
    outr[i] = f2b(b2f(xr[i]) * scale * b2f(w[i]));
}

// ---------------------------------------------------------------------------
// silu_mul (ops.cu:101-138): out = gate * sigmoid(gate) * up
// ---------------------------------------------------------------------------
__global__ void k_silu_mul(const u16* __restrict__ gate,
                           const u16* __restrict__ up, u16* __restrict__ out,
                           size_t n) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float g = b2f(gate[i]);
    float u = b2f(up[i]);
    out[i] = f2b(g / (1.f + __expf(-g)) * u);
  }
}
// strided variant for the fused gate_up buffer (S, 2I): gate = row[0:I],
// up = row[I:2I]  (mlp.rs:21-31 narrow semantics)
__global__ void k_silu_mul_rows(const u16* __restrict__ gu,
                                u16* __restrict__ out, int S, int I) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  size_t n = (size_t)S * I;
  for (; i < n; i += stride) {
    int s = (int)(i / I), c = (int)(i % I);
    float g = b2f(gu[(size_t)s * 2 * I + c]);
    float u = b2f(gu[(size_t)s * 2 * I + I + c]);
    out[i] = f2b(g / (1.f + __expf(-g)) * u);
  }
}

// ---------------------------------------------------------------------------
// GEMV family — decode's workhorse (HBM-bound weight streaming at 16 B/lane,
// guide §5 "GEMV / M <= 16": weights straight to VGPRs, no LDS round trip).
//
// k_gemv_reg<ROWS, EPI, NORM, KB>: x is loaded ONCE into registers (up to
// KB*8 f32/thread, K <= KB*2048) and reused for every row; with NORM the
// kernel fuses the preceding rms_norm (backends/mod.rs:244-246) into the
// x load — the normed value is re-quantized to bf16 so the fused path is
// bit-identical to rmsnorm-then-gemv.  ROWS per block is chosen by the
// launcher so the grid has >= ~2048 workgroups (256 CUs want many blocks).
// EPI: 0 = bf16 out; 1 = bf16 out + residual add; 2 = f32 out (logits).
// ---------------------------------------------------------------------------
template <int ROWS, int EPI, bool NORM, int KB>
__global__ __launch_bounds__(256) void k_gemv_reg(
    const u16* __restrict__ W, const u16* __restrict__ x,
    void* __restrict__ out, const u16* __restrict__ res,
    const u16* __restrict__ nw, float eps, int N, int K) {
  const int t = threadIdx.x;
  const int row0 = blockIdx.x * ROWS;
  const int wid = t / WAVE, lane = t % WAVE;
  __shared__ float red[ROWS > 4 ? ROWS : 4][4];

  // For short K, issue ALL weight loads first — they stay in flight across
  // the x/norm phase (plain VGPR loads survive s_barrier; guide §5
  // pipelining note), hiding the norm reduction latency entirely.
  short8 wpre[KB <= 2 ? ROWS : 1][KB <= 2 ? KB : 1];
  if (KB <= 2) {
#pragma unroll
    for (int r = 0; r < ROWS; ++r)
#pragma unroll
      for (int i = 0; i < KB; ++i) {
        const int k0 = i * 2048 + t * 8;
        if (row0 + r < N && k0 < K)
          wpre[r][i] = ntload8(W + (size_t)(row0 + r) * K + k0);
      }
  }

  // phase 1: x -> registers (f32), optionally fused rms_norm
  float xr[KB * 8];
#pragma unroll
  for (int i = 0; i < KB; ++i) {
    const int k0 = i * 2048 + t * 8;
    if (k0 < K) {
      short8 xv = *reinterpret_cast<const short8*>(x + k0);
#pragma unroll
      for (int j = 0; j < 8; ++j) xr[i * 8 + j] = b2f((u16)xv[j]);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) xr[i * 8 + j] = 0.f;
    }
  }
  if (NORM) {
    float ss = 0.f;
#pragma unroll
    for (int i = 0; i < KB * 8; ++i) ss += xr[i] * xr[i];
    ss = wave_sum(ss);
    if (lane == 0) red[0][wid] = ss;
    __syncthreads();
    const float scale =
        rsqrtf((red[0][0] + red[0][1] + red[0][2] + red[0][3]) / (float)K +
               eps);
    __syncthreads();
#pragma unroll
    for (int i = 0; i < KB; ++i) {
      const int k0 = i * 2048 + t * 8;
      if (k0 < K) {
        short8 wv = *reinterpret_cast<const short8*>(nw + k0);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          // re-quantize so fused == rmsnorm-kernel-then-gemv bit-exactly
          xr[i * 8 + j] = b2f(f2b(xr[i * 8 + j] * scale * b2f((u16)wv[j])));
      }
    }
  }

  // phase 2: FMA with the prefetched weights (KB<=2) or stream rows with
  // all rows' loads unrolled together (KB>2) — memory-level parallelism
  float acc[ROWS];
#pragma unroll
  for (int r = 0; r < ROWS; ++r) acc[r] = 0.f;
  if (KB <= 2) {
#pragma unroll
    for (int r = 0; r < ROWS; ++r)
#pragma unroll
      for (int i = 0; i < KB; ++i) {
        const int k0 = i * 2048 + t * 8;
        if (row0 + r < N && k0 < K) {
#pragma unroll
          for (int j = 0; j < 8; ++j)
            acc[r] = fmaf(b2f((u16)wpre[r][i][j]), xr[i * 8 + j], acc[r]);
        }
      }
  } else {
#pragma unroll
    for (int i = 0; i < KB; ++i) {
      const int k0 = i * 2048 + t * 8;
      if (k0 < K) {
        short8 wv[ROWS];
#pragma unroll
        for (int r = 0; r < ROWS; ++r)
          if (row0 + r < N) wv[r] = ntload8(W + (size_t)(row0 + r) * K + k0);
#pragma unroll
        for (int r = 0; r < ROWS; ++r)
          if (row0 + r < N) {
#pragma unroll
            for (int j = 0; j < 8; ++j)
              acc[r] = fmaf(b2f((u16)wv[r][j]), xr[i * 8 + j], acc[r]);
          }
      }
    }
  }
#pragma unroll
  for (int r = 0; r < ROWS; ++r) {
    float v = wave_sum(acc[r]);
    if (lane == 0) red[r][wid] = v;
  }
  __syncthreads();
  if (t < ROWS) {
    const int row = row0 + t;
    if (row < N) {
      float v = red[t][0] + red[t][1] + red[t][2] + red[t][3];
      if (EPI == 2) {
        reinterpret_cast<float*>(out)[row] = v;
      } else if (EPI == 1) {
        reinterpret_cast<u16*>(out)[row] = f2b(v + b2f(res[row]));
      } else {
        reinterpret_cast<u16*>(out)[row] = f2b(v);
      }
    }
  }
}

// streaming fallback for K > 16384 (e.g. 70B down-proj K=28672): x re-read
// from L1/L2 per row
template <int ROWS, int EPI>
__global__ __launch_bounds__(256) void k_gemv_stream(
    const u16* __restrict__ W, const u16* __restrict__ x,
    void* __restrict__ out, const u16* __restrict__ res, int N, int K) {
  const int t = threadIdx.x;
  const int row0 = blockIdx.x * ROWS;
  const int wid = t / WAVE, lane = t % WAVE;
  float acc[ROWS];
#pragma unroll
  for (int r = 0; r < ROWS; ++r) acc[r] = 0.f;
  const int kiter = (K + 2047) / 2048;
#pragma unroll 1
  for (int r = 0; r < ROWS; ++r) {
    const int row = row0 + r;
    if (row >= N) break;
    const u16* wr = W + (size_t)row * K;
    float a = 0.f;
    for (int i = 0; i < kiter; ++i) {
      int k0 = i * 2048 + t * 8;
      if (k0 + 8 <= K) {
        short8 wv = ntload8(wr + k0);
        short8 xv = *reinterpret_cast<const short8*>(x + k0);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          a = fmaf(b2f((u16)wv[j]), b2f((u16)xv[j]), a);
      } else {
        for (int k = k0; k < K; ++k) a = fmaf(b2f(wr[k]), b2f(x[k]), a);
      }
    }
    acc[r] = a;
  }
  __shared__ float red[ROWS > 4 ? ROWS : 4][4];
#pragma unroll
  for (int r = 0; r < ROWS; ++r) {
    float v = wave_sum(acc[r]);
    if (lane == 0) red[r][wid] = v;
  }
  __syncthreads();
  if (t < ROWS) {
    const int row = row0 + t;
    if (row < N) {
      float v = red[t][0] + red[t][1] + red[t][2] + red[t][3];
      if (EPI == 2) {
        reinterpret_cast<float*>(out)[row] = v;
      } else if (EPI == 1) {
        reinterpret_cast<u16*>(out)[row] = f2b(v + b2f(res[row]));
      } else {
        reinterpret_cast<u16*>(out)[row] = f2b(v);
      }
    }
  }
}

// fused [rms_norm ->] gate_up GEMV -> silu_mul: block computes
// out[i] = silu(g_i) * u_i for 8 channels, g_i = dot(W[i,:], xn),
// u_i = dot(W[i+I,:], xn)  (mlp.rs:21-31 + fused gate_up of mlp.rs:38-46)
template <int ROWS, bool NORM, int KB>
__global__ __launch_bounds__(256) void k_gemv_gateup(
    const u16* __restrict__ W, const u16* __restrict__ x,
    u16* __restrict__ out, const u16* __restrict__ nw, float eps, int I,
    int K) {
  const int t = threadIdx.x;
  const int c0 = blockIdx.x * ROWS;
  const int wid = t / WAVE, lane = t % WAVE;
  __shared__ float redg[8][4], redu[8][4];

  float xr[KB * 8];
#pragma unroll
  for (int i = 0; i < KB; ++i) {
    const int k0 = i * 2048 + t * 8;
    if (k0 < K) {
      short8 xv = *reinterpret_cast<const short8*>(x + k0);
#pragma unroll
      for (int j = 0; j < 8; ++j) xr[i * 8 + j] = b2f((u16)xv[j]);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) xr[i * 8 + j] = 0.f;
    }
  }
  if (NORM) {
    float ss = 0.f;
#pragma unroll
    for (int i = 0; i < KB * 8; ++i) ss += xr[i] * xr[i];
    ss = wave_sum(ss);
    if (lane == 0) redg[0][wid] = ss;
    __syncthreads();
    const float scale =
        rsqrtf((redg[0][0] + redg[0][1] + redg[0][2] + redg[0][3]) /
                   (float)K + eps);
    __syncthreads();
#pragma unroll
    for (int i = 0; i < KB; ++i) {
      const int k0 = i * 2048 + t * 8;
      if (k0 < K) {
        short8 wv = *reinterpret_cast<const short8*>(nw + k0);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          xr[i * 8 + j] = b2f(f2b(xr[i * 8 + j] * scale * b2f((u16)wv[j])));
      }
    }
  }

  float accg[ROWS], accu[ROWS];
#pragma unroll
  for (int r = 0; r < ROWS; ++r) accg[r] = accu[r] = 0.f;
#pragma unroll
  for (int i = 0; i < KB; ++i) {
    const int k0 = i * 2048 + t * 8;
    if (k0 < K) {
      short8 gv[ROWS], uv[ROWS];
#pragma unroll
      for (int r = 0; r < ROWS; ++r)
        if (c0 + r < I) {
          gv[r] = ntload8(W + (size_t)(c0 + r) * K + k0);
          uv[r] = ntload8(W + (size_t)(c0 + r + I) * K + k0);
        }
#pragma unroll
      for (int r = 0; r < ROWS; ++r)
        if (c0 + r < I) {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            accg[r] = fmaf(b2f((u16)gv[r][j]), xr[i * 8 + j], accg[r]);
            accu[r] = fmaf(b2f((u16)uv[r][j]), xr[i * 8 + j], accu[r]);
          }
        }
    }
  }
#pragma unroll
  for (int r = 0; r < ROWS; ++r) {
    float g = wave_sum(accg[r]);
    float u = wave_sum(accu[r]);
    if (lane == 0) { redg[r][wid] = g; redu[r][wid] = u; }
  }
  __syncthreads();
  if (t < ROWS && c0 + t < I) {
    float g = redg[t][0] + redg[t][1] + redg[t][2] + redg[t][3];
    float u = redu[t][0] + redu[t][1] + redu[t][2] + redu[t][3];
    out[c0 + t] = f2b(g / (1.f + __expf(-g)) * u);
  }
}

// ---------------------------------------------------------------------------
// FP8 GEMV family — same structure as k_gemv_reg but weights are e4m3fn
// bytes with blockwise 128x128 scale_inv (utils/fp8.rs:42-64), dequantized
// in-register: HALF the HBM bytes per decode step.  16 weights per 16-B
// lane load.  KB here = ceil(K/4096) (thread covers 16 elems per iter).
// ---------------------------------------------------------------------------
template <int ROWS, int EPI, bool NORM, int KB>
__global__ __launch_bounds__(256) void k_gemv_fp8(
    const unsigned char* __restrict__ W, const float* __restrict__ sc,
    const u16* __restrict__ x, void* __restrict__ out,
    const u16* __restrict__ res, const u16* __restrict__ nw, float eps,
    int N, int K, int nkb) {
  const int t = threadIdx.x;
  const int row0 = blockIdx.x * ROWS;
  const int wid = t / WAVE, lane = t % WAVE;
  __shared__ float red[ROWS > 4 ? ROWS : 4][4];

  // prefetch all weight tiles (stay in flight across the x/norm phase)
  uint4v wpre[ROWS][KB];
#pragma unroll
  for (int r = 0; r < ROWS; ++r)
#pragma unroll
    for (int i = 0; i < KB; ++i) {
      const int k0 = i * 4096 + t * 16;
      if (row0 + r < N && k0 < K)
        wpre[r][i] = ntload16b(W + (size_t)(row0 + r) * K + k0);
    }

  // x -> registers (f32), optionally fused rms_norm
  float xr[KB * 16];
#pragma unroll
  for (int i = 0; i < KB; ++i) {
    const int k0 = i * 4096 + t * 16;
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      if (k0 < K) {
        short8 xv = *reinterpret_cast<const short8*>(x + k0 + half * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          xr[i * 16 + half * 8 + j] = b2f((u16)xv[j]);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) xr[i * 16 + half * 8 + j] = 0.f;
      }
    }
  }
  if (NORM) {
    float ss = 0.f;
#pragma unroll
    for (int i = 0; i < KB * 16; ++i) ss += xr[i] * xr[i];
    ss = wave_sum(ss);
    if (lane == 0) red[0][wid] = ss;
    __syncthreads();
    const float scale =
        rsqrtf((red[0][0] + red[0][1] + red[0][2] + red[0][3]) / (float)K +
               eps);
    __syncthreads();
#pragma unroll
    for (int i = 0; i < KB; ++i) {
      const int k0 = i * 4096 + t * 16;
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        if (k0 < K) {
          short8 wv =
              *reinterpret_cast<const short8*>(nw + k0 + half * 8);
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            const int ix = i * 16 + half * 8 + j;
            xr[ix] = b2f(f2b(xr[ix] * scale * b2f((u16)wv[j])));
          }
        }
      }
    }
  }

  float acc[ROWS];
#pragma unroll
  for (int r = 0; r < ROWS; ++r) acc[r] = 0.f;
#pragma unroll
  for (int r = 0; r < ROWS; ++r)
#pragma unroll
    for (int i = 0; i < KB; ++i) {
      const int k0 = i * 4096 + t * 16;
      if (row0 + r < N && k0 < K) {
        // one scale_inv block covers this thread's 16 k's (k0 % 16 == 0)
        const float s =
            sc[(size_t)((row0 + r) / 128) * nkb + (k0 / 128)];
        float wd[16];
        f8x16_decode(wpre[r][i], wd);
        float a = 0.f;
#pragma unroll
        for (int j = 0; j < 16; ++j)
          a = fmaf(wd[j], xr[i * 16 + j], a);
        acc[r] = fmaf(a, s, acc[r]);
      }
    }
#pragma unroll
  for (int r = 0; r < ROWS; ++r) {
    float v = wave_sum(acc[r]);
    if (lane == 0) red[r][wid] = v;
  }
  __syncthreads();
  if (t < ROWS) {
    const int row = row0 + t;
    if (row < N) {
      float v = red[t][0] + red[t][1] + red[t][2] + red[t][3];
      if (EPI == 2) {
        reinterpret_cast<float*>(out)[row] = v;
      } else if (EPI == 1) {
        reinterpret_cast<u16*>(out)[row] = f2b(v + b2f(res[row]));
      } else {
        reinterpret_cast<u16*>(out)[row] = f2b(v);
      }
    }
  }
}

// fp8 streaming fallback for K > 16384 (e.g. Qwen3-32B down K=25600)
template <int ROWS, int EPI>
__global__ __launch_bounds__(256) void k_gemv_fp8_stream(
    const unsigned char* __restrict__ W, const float* __restrict__ sc,
    const u16* __restrict__ x, void* __restrict__ out,
    const u16* __restrict__ res, int N, int K, int nkb) {
  const int t = threadIdx.x;
  const int row0 = blockIdx.x * ROWS;
  const int wid = t / WAVE, lane = t % WAVE;
  float acc[ROWS];
#pragma unroll
  for (int r = 0; r < ROWS; ++r) acc[r] = 0.f;
  const int kiter = (K + 4095) / 4096;
#pragma unroll 1
  for (int i = 0; i < kiter; ++i) {
    const int k0 = i * 4096 + t * 16;
    if (k0 >= K) continue;
    float xv[16];
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      short8 xs = *reinterpret_cast<const short8*>(x + k0 + half * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) xv[half * 8 + j] = b2f((u16)xs[j]);
    }
    uint4v wv[ROWS];
#pragma unroll
    for (int r = 0; r < ROWS; ++r)
      if (row0 + r < N) wv[r] = ntload16b(W + (size_t)(row0 + r) * K + k0);
#pragma unroll
    for (int r = 0; r < ROWS; ++r)
      if (row0 + r < N) {
        const float s = sc[(size_t)((row0 + r) / 128) * nkb + (k0 / 128)];
        float wd[16];
        f8x16_decode(wv[r], wd);
        float a = 0.f;
#pragma unroll
        for (int j = 0; j < 16; ++j)
          a = fmaf(wd[j], xv[j], a);
        acc[r] = fmaf(a, s, acc[r]);
      }
  }
  __shared__ float red[ROWS > 4 ? ROWS : 4][4];
#pragma unroll
  for (int r = 0; r < ROWS; ++r) {
    float v = wave_sum(acc[r]);
    if (lane == 0) red[r][wid] = v;
  }
  __syncthreads();
  if (t < ROWS) {
    const int row = row0 + t;
    if (row < N) {
      float v = red[t][0] + red[t][1] + red[t][2] + red[t][3];
      if (EPI == 2) {
        reinterpret_cast<float*>(out)[row] = v;
      } else if (EPI == 1) {
        reinterpret_cast<u16*>(out)[row] = f2b(v + b2f(res[row]));
      } else {
        reinterpret_cast<u16*>(out)[row] = f2b(v);
      }
    }
  }
}

// fp8 gate_up + silu_mul (the fused MLP front half, fp8 weights)
template <int ROWS, bool NORM, int KB>
__global__ __launch_bounds__(256) void k_gemv_gateup_fp8(
    const unsigned char* __restrict__ W, const float* __restrict__ sc,
    const u16* __restrict__ x, u16* __restrict__ out,
    const u16* __restrict__ nw, float eps, int I, int K, int nkb) {
  const int t = threadIdx.x;
  const int c0 = blockIdx.x * ROWS;
  const int wid = t / WAVE, lane = t % WAVE;
  __shared__ float redg[8][4], redu[8][4];

  float xr[KB * 16];
#pragma unroll
  for (int i = 0; i < KB; ++i) {
    const int k0 = i * 4096 + t * 16;
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      if (k0 < K) {
        short8 xv = *reinterpret_cast<const short8*>(x + k0 + half * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          xr[i * 16 + half * 8 + j] = b2f((u16)xv[j]);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) xr[i * 16 + half * 8 + j] = 0.f;
      }
    }
  }
  if (NORM) {
    float ss = 0.f;
#pragma unroll
    for (int i = 0; i < KB * 16; ++i) ss += xr[i] * xr[i];
    ss = wave_sum(ss);
    if (lane == 0) redg[0][wid] = ss;
    __syncthreads();
    const float scale =
        rsqrtf((redg[0][0] + redg[0][1] + redg[0][2] + redg[0][3]) /
                   (float)K + eps);
    __syncthreads();
#pragma unroll
    for (int i = 0; i < KB; ++i) {
      const int k0 = i * 4096 + t * 16;
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        if (k0 < K) {
          short8 wv =
              *reinterpret_cast<const short8*>(nw + k0 + half * 8);
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            const int ix = i * 16 + half * 8 + j;
            xr[ix] = b2f(f2b(xr[ix] * scale * b2f((u16)wv[j])));
          }
        }
      }
    }
  }

  float accg[ROWS], accu[ROWS];
#pragma unroll
  for (int r = 0; r < ROWS; ++r) accg[r] = accu[r] = 0.f;
#pragma unroll
  for (int i = 0; i < KB; ++i) {
    const int k0 = i * 4096 + t * 16;
    if (k0 >= K) continue;
    uint4v gv[ROWS], uv[ROWS];
#pragma unroll
    for (int r = 0; r < ROWS; ++r)
      if (c0 + r < I) {
        gv[r] = ntload16b(W + (size_t)(c0 + r) * K + k0);
        uv[r] = ntload16b(W + (size_t)(c0 + r + I) * K + k0);
      }
#pragma unroll
    for (int r = 0; r < ROWS; ++r)
      if (c0 + r < I) {
        const float sg = sc[(size_t)((c0 + r) / 128) * nkb + (k0 / 128)];
        const float su =
            sc[(size_t)((c0 + r + I) / 128) * nkb + (k0 / 128)];
        float gd[16], ud[16];
        f8x16_decode(gv[r], gd);
        f8x16_decode(uv[r], ud);
        float g = 0.f, u = 0.f;
#pragma unroll
        for (int j = 0; j < 16; ++j) {
          g = fmaf(gd[j], xr[i * 16 + j], g);
          u = fmaf(ud[j], xr[i * 16 + j], u);
        }
        accg[r] = fmaf(g, sg, accg[r]);
        accu[r] = fmaf(u, su, accu[r]);
      }
  }
#pragma unroll
  for (int r = 0; r < ROWS; ++r) {
    float g = wave_sum(accg[r]);
    float u = wave_sum(accu[r]);
    if (lane == 0) { redg[r][wid] = g; redu[r][wid] = u; }
  }
  __syncthreads();
  if (t < ROWS && c0 + t < I) {
    float g = redg[t][0] + redg[t][1] + redg[t][2] + redg[t][3];
    float u = redu[t][0] + redu[t][1] + redu[t][2] + redu[t][3];
    out[c0 + t] = f2b(g / (1.f + __expf(-g)) * u);
  }
}

// fp8 -> bf16 blockwise dequant (prefill path: dequant the layer's weight
// into a scratch buffer, then run the bf16 MFMA GEMM — cake itself
// dequantizes at load time, fp8.rs:42-64; we dequant per layer per prefill
// to keep HBM fp8-resident for decode)
__global__ void k_dequant_fp8(const unsigned char* __restrict__ W,
                              const float* __restrict__ sc,
                              u16* __restrict__ out, int N, int K, int nkb) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  size_t total = (size_t)N * K;
  for (; i < total; i += stride) {
    int r = (int)(i / K), k = (int)(i % K);
    f32x2 v = __builtin_amdgcn_cvt_pk_f32_fp8((u32)W[i], false);
    out[i] = f2b(v[0] * sc[(size_t)(r / 128) * nkb + k / 128]);
  }
}

// ---------------------------------------------------------------------------
// embedding gather (backends/mod.rs:513-528)
// ---------------------------------------------------------------------------
__global__ void k_embed_token(const u16* __restrict__ embed,
                              const u32* __restrict__ tok,
                              u16* __restrict__ x, int H) {
  const u16* src = embed + (size_t)(*tok) * H;
  const int nv = H / 8;
  for (int i = threadIdx.x; i < nv; i += blockDim.x)
    *reinterpret_cast<short8*>(x + i * 8) =
        *reinterpret_cast<const short8*>(src + i * 8);
  for (int i = nv * 8 + threadIdx.x; i < H; i += blockDim.x) x[i] = src[i];
}
__global__ void k_embed_rows(const u16* __restrict__ embed,
                             const u32* __restrict__ ids,
                             u16* __restrict__ x, int H) {
  const int s = blockIdx.x;
  const u16* src = embed + (size_t)ids[s] * H;
  u16* dst = x + (size_t)s * H;
  for (int i = threadIdx.x; i < H; i += blockDim.x) dst[i] = src[i];
}

// ---------------------------------------------------------------------------
// RoPE + KV store.  HF half-rotation (backends/mod.rs:470-477):
//   out[i] = x1*c - x2*s ; out[i+half] = x2*c + x1*s, cos/sin row = position.
// Decode: one token at *pos; K/V written into the preallocated cache at slot
// *pos (replaces cache.rs:195-196 cat).  qkv layout: (Sq | Skv | Skv) per row.
// grid = nh + 2*nkv blocks.
// ---------------------------------------------------------------------------
// Optional fused per-head QK-norm (attention.rs:202-215): applied to the
// head row before the rotation, re-quantized to bf16 so it matches the
// separate rmsnorm-kernel path bit-exactly.  One wave per head.
__device__ inline float head_norm_scale(const u16* row, int hd, float eps) {
  float ss = 0.f;
  for (int i = threadIdx.x; i < hd; i += 64) {
    float f = b2f(row[i]);
    ss += f * f;
  }
  ss = wave_sum(ss);
  ss = __shfl(ss, 0, WAVE);
  return rsqrtf(ss / (float)hd + eps);
}
__device__ inline float nrm(const u16* row, const u16* w, int i, float sc,
                            bool on) {
  float v = b2f(row[i]);
  return on ? b2f(f2b(v * sc * b2f(w[i]))) : v;
}

__global__ void k_rope_store_decode(u16* __restrict__ qkv,
                                    u16* __restrict__ kc, u16* __restrict__ vc,
                                    u16* __restrict__ vtc,
                                    const float* __restrict__ cost,
                                    const float* __restrict__ sint,
                                    const int* __restrict__ pos, int nh,
                                    int nkv, int hd, int rd, int max_seq,
                                    const u16* __restrict__ qn,
                                    const u16* __restrict__ kn, float eps) {
  const int b = blockIdx.x;
  const int p = *pos;
  const int half = rd / 2;
  const float* c = cost + (size_t)p * half;
  const float* s = sint + (size_t)p * half;
  if (b < nh) {                       // [qk-norm +] rope q head in place
    u16* q = qkv + (size_t)b * hd;
    const bool on = qn != nullptr;
    const float sc = on ? head_norm_scale(q, hd, eps) : 1.f;
    for (int i = threadIdx.x; i < half; i += blockDim.x) {
      float x1 = nrm(q, qn, i, sc, on), x2 = nrm(q, qn, i + half, sc, on);
      q[i] = f2b(x1 * c[i] - x2 * s[i]);
      q[i + half] = f2b(x2 * c[i] + x1 * s[i]);
    }
  } else if (b < nh + nkv) {          // [qk-norm +] rope k head -> slot p
    const int h = b - nh;
    u16* k = qkv + (size_t)(nh + h) * hd;
    u16* dst = kc + ((size_t)h * max_seq + p) * hd;
    const bool on = kn != nullptr;
    const float sc = on ? head_norm_scale(k, hd, eps) : 1.f;
    for (int i = threadIdx.x; i < half; i += blockDim.x) {
      float x1 = nrm(k, kn, i, sc, on), x2 = nrm(k, kn, i + half, sc, on);
      dst[i] = f2b(x1 * c[i] - x2 * s[i]);
      dst[i + half] = f2b(x2 * c[i] + x1 * s[i]);
    }
    for (int i = rd + threadIdx.x; i < hd; i += blockDim.x) dst[i] = k[i];
  } else {                            // v head -> cache slot p (+ V^T)
    const int h = b - nh - nkv;
    const u16* v = qkv + (size_t)(nh + nkv + h) * hd;
    u16* dst = vc + ((size_t)h * max_seq + p) * hd;
    u16* dstt = vtc + (size_t)h * hd * max_seq + p;
    for (int i = threadIdx.x; i < hd; i += blockDim.x) {
      dst[i] = v[i];
      dstt[(size_t)i * max_seq] = v[i];
    }
  }
}

// Prefill: S tokens at positions pos0..pos0+S-1; qkv is (S, Sq+2*Skv).
// grid = (nh + 2*nkv, S)
__global__ void k_rope_store_prefill(u16* __restrict__ qkv,
                                     u16* __restrict__ kc, u16* __restrict__ vc,
                                     u16* __restrict__ vtc,
                                     const float* __restrict__ cost,
                                     const float* __restrict__ sint, int pos0,
                                     int nh, int nkv, int hd, int rd,
                                     int max_seq, int qkv_stride,
                                     const u16* __restrict__ qn,
                                     const u16* __restrict__ kn, float eps) {
  const int b = blockIdx.x;
  const int sidx = blockIdx.y;
  const int p = pos0 + sidx;
  const int half = rd / 2;
  const float* c = cost + (size_t)p * half;
  const float* s = sint + (size_t)p * half;
  u16* row = qkv + (size_t)sidx * qkv_stride;
  if (b < nh) {
    u16* q = row + (size_t)b * hd;
    const bool on = qn != nullptr;
    const float sc = on ? head_norm_scale(q, hd, eps) : 1.f;
    for (int i = threadIdx.x; i < half; i += blockDim.x) {
      float x1 = nrm(q, qn, i, sc, on), x2 = nrm(q, qn, i + half, sc, on);
      q[i] = f2b(x1 * c[i] - x2 * s[i]);
      q[i + half] = f2b(x2 * c[i] + x1 * s[i]);
    }
  } else if (b < nh + nkv) {
    const int h = b - nh;
    u16* k = row + (size_t)(nh + h) * hd;
    u16* dst = kc + ((size_t)h * max_seq + p) * hd;
    const bool on = kn != nullptr;
    const float sc = on ? head_norm_scale(k, hd, eps) : 1.f;
    for (int i = threadIdx.x; i < half; i += blockDim.x) {
      float x1 = nrm(k, kn, i, sc, on), x2 = nrm(k, kn, i + half, sc, on);
      dst[i] = f2b(x1 * c[i] - x2 * s[i]);
      dst[i + half] = f2b(x2 * c[i] + x1 * s[i]);
    }
    for (int i = rd + threadIdx.x; i < hd; i += blockDim.x) dst[i] = k[i];
  } else {
    const int h = b - nh - nkv;
    const u16* v = row + (size_t)(nh + nkv + h) * hd;
    u16* dst = vc + ((size_t)h * max_seq + p) * hd;
    u16* dstt = vtc + (size_t)h * hd * max_seq + p;
    for (int i = threadIdx.x; i < hd; i += blockDim.x) {
      dst[i] = v[i];
      dstt[(size_t)i * max_seq] = v[i];
    }
  }
}

// ---------------------------------------------------------------------------
// Decode attention (attention.rs:300-343 semantics, GQA, f32 softmax), over
// the preallocated cache.  Split-KV in ONE launch: grid (NCHUNK, nh); each
// block computes an online-softmax partial over a contiguous slice of
// positions into ws[h][chunk] = {o[hd], m, l}; the LAST-arriving block of a
// head combines the partials (agent-scope release/acquire + arrival counter
// per cdna_hip_programming.md §6 Guideline 16 — placement-independent).
// cnt[nh] must be zeroed before every launch (hipMemsetAsync node).
// K/V row loads: 64 lanes x 4 B = one coalesced 256 B transaction.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_attn_decode_fused(
    const u16* __restrict__ q,         // (nh*hd), post-rope
    const u16* __restrict__ kc, const u16* __restrict__ vc,
    const int* __restrict__ pos, float* __restrict__ ws,
    u32* __restrict__ cnt, u16* __restrict__ outbuf, int nh, int nkv,
    int hd, int max_seq, int nchunk) {
  const int h = blockIdx.y;
  const int chunk = blockIdx.x;
  const int n = *pos + 1;
  const int cs = (n + nchunk - 1) / nchunk;
  const int start = chunk * cs;
  const int end = min(start + cs, n);
  const int kvh = h / (nh / nkv);
  const int t = threadIdx.x, wid = t / WAVE, lane = t % WAVE;
  const int e0 = 2 * lane;             // dims (2*lane, 2*lane+1)
  const bool act = e0 + 1 < hd;
  float* wsrow = ws + ((size_t)h * nchunk + chunk) * (hd + 2);
#define WS_STORE(p, v)                                                     \
  __hip_atomic_store((p), (v), __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT)
#define WS_LOAD(p)                                                         \
  __hip_atomic_load((p), __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT)

  // Tiled two-phase structure: the serial online-softmax chain runs once
  // per TILE (128 positions), not per position — phase A computes a tile's
  // scores with 16-B coalesced K loads (16 lanes per position), phase B is
  // a block-wide softmax step, phase C accumulates PV with coalesced,
  // mutually independent V loads.  The per-position serial chain of the
  // naive version measured 128 GB/s at n=2048.
  __shared__ float sm[4], sl[4];
  __shared__ float so[4][128 + 8];
  __shared__ float stile[128];
  const float scale = rsqrtf((float)hd);
  const u16* kbase = kc + (size_t)kvh * max_seq * hd;
  const u16* vbase = vc + (size_t)kvh * max_seq * hd;
  // phase-A per-thread q slice: dims (t&15)*8 .. +8
  const int dgrp = t & 15;
  float qa[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int d = dgrp * 8 + j;
    qa[j] = d < hd ? b2f(q[(size_t)h * hd + d]) : 0.f;
  }
  float m = -INFINITY, l = 0.f, o0 = 0.f, o1 = 0.f;
  const int TILE = 128;
  for (int sub0 = start; sub0 < end; sub0 += TILE) {
    // --- phase A: scores for [sub0, sub0+TILE) --------------------------
    // 256 threads = 16 positions per pass (16 lanes per position, each
    // loading 16 B of the K row)
#pragma unroll 4
    for (int pass = 0; pass < TILE / 16; ++pass) {
      const int p = sub0 + pass * 16 + (t >> 4);
      float d = 0.f;
      if (p < end && dgrp * 8 < hd) {
        const u16* kr = kbase + (size_t)p * hd + dgrp * 8;
        short8 kv8 = *reinterpret_cast<const short8*>(kr);
#pragma unroll
        for (int j = 0; j < 8; ++j) d = fmaf(b2f((u16)kv8[j]), qa[j], d);
      }
      // reduce across the 16 lanes of this position
#pragma unroll
      for (int off = 8; off >= 1; off >>= 1) d += __shfl_xor(d, off, 16);
      if (dgrp == 0)
        stile[pass * 16 + (t >> 4)] = (p < end) ? d * scale : -INFINITY;
    }
    __syncthreads();
    // --- phase B: block softmax step over the tile ----------------------
    float lm = -INFINITY;
    for (int i = t; i < TILE; i += blockDim.x) lm = fmaxf(lm, stile[i]);
    lm = wave_max(lm);
    if (lane == 0) sm[wid] = lm;
    __syncthreads();
    const float tmax = fmaxf(fmaxf(sm[0], sm[1]), fmaxf(sm[2], sm[3]));
    const float mnew = fmaxf(m, tmax);
    const float alpha = (mnew == -INFINITY) ? 0.f : __expf(m - mnew);
    float psum = 0.f;
    for (int i = t; i < TILE; i += blockDim.x) {
      const float sv = stile[i];
      const float e = (sv == -INFINITY) ? 0.f : __expf(sv - mnew);
      stile[i] = e;
      psum += e;
    }
    psum = wave_sum(psum);
    if (lane == 0) sl[wid] = psum;
    __syncthreads();
    l = l * alpha + sl[0] + sl[1] + sl[2] + sl[3];
    o0 *= alpha;
    o1 *= alpha;
    m = mnew;
    // --- phase C: PV accumulate (thread t: dims 2*lane, position residue
    // wid mod 4; loads independent across iterations) --------------------
#pragma unroll 8
    for (int k = 0; k < TILE / 4; ++k) {
      const int po = k * 4 + wid;
      const int p = sub0 + po;
      if (act && p < end) {
        const float w = stile[po];
        const u16* vr = vbase + (size_t)p * hd + e0;
        o0 = fmaf(w, b2f(vr[0]), o0);
        o1 = fmaf(w, b2f(vr[1]), o1);
      }
    }
    __syncthreads();  // stile reused next sub-tile
  }
  // combine the 4 position-residue partials per dim
  if (act) { so[wid][e0] = o0; so[wid][e0 + 1] = o1; }
  __syncthreads();
  if (t == 0) {
    WS_STORE(&wsrow[hd], m);
    WS_STORE(&wsrow[hd + 1], l);
  }
  for (int d = t; d < hd; d += blockDim.x)
    WS_STORE(&wsrow[d], so[0][d] + so[1][d] + so[2][d] + so[3][d]);

  // ---- publish partial + elect the combining block ----------------------
  // sc1 write-through publish (Guideline 16 R1 variant): the ws stores above
  // are agent-scope relaxed 4-B atomics (= sc1 stores, the natural width
  // here), so no release fence is needed — just drain, then count arrivals.
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // every wave drains
  __syncthreads();
  if (t == 0) {
    // epoch-free election: counters monotonically accumulate, the block
    // drawing (v % nchunk) == nchunk-1 combines — no per-launch reset
    u32 v = __hip_atomic_fetch_add(&cnt[h], 1u, __ATOMIC_RELAXED,
                                   __HIP_MEMORY_SCOPE_AGENT);
    sm[0] = (v % (u32)nchunk == (u32)(nchunk - 1)) ? 1.f : 0.f;
  }
  __syncthreads();
  if (sm[0] == 0.f) return;
  // reducer reads the slabs with sc1 loads — no acquire fence, no L1 risk

  // ---- combine this head's partials (runs in exactly one block) ----------
  float* base = ws + (size_t)h * nchunk * (hd + 2);
  // stage m,l in LDS (parallel sc1 loads; serial dependent uncached loads
  // were the reducer's cost), then combine with per-chunk weights from LDS
  if (t < nchunk) {
    sm[0] = 0.f;  // keep sm[0] clear; use so rows as staging
    so[0][t] = WS_LOAD(&base[t * (hd + 2) + hd]);
    so[1][t] = WS_LOAD(&base[t * (hd + 2) + hd + 1]);
  }
  __syncthreads();
  float M = -INFINITY;
  for (int c = 0; c < nchunk; ++c) M = fmaxf(M, so[0][c]);
  float L = 0.f;
  for (int c = 0; c < nchunk; ++c)
    if (so[0][c] != -INFINITY) L += so[1][c] * __expf(so[0][c] - M);
  for (int d = t; d < hd; d += blockDim.x) {
    float o = 0.f;
#pragma unroll 4
    for (int c = 0; c < nchunk; ++c)
      if (so[0][c] != -INFINITY)
        o += WS_LOAD(&base[c * (hd + 2) + d]) * __expf(so[0][c] - M);
    outbuf[(size_t)h * hd + d] = f2b(o / L);
  }
#undef WS_STORE
#undef WS_LOAD
}

// ---------------------------------------------------------------------------
// Prefill attention — flash-style f32 online softmax, causal, GQA, over the
// cache (which already holds positions [0, pos0+S)).  One wave per query
// row; 4 rows per block.  (attention.rs:300-343 + cache.rs:150-160 mask.)
// q rows come from the post-rope qkv buffer.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_attn_prefill(
    const u16* __restrict__ qkv, const u16* __restrict__ kc,
    const u16* __restrict__ vc, u16* __restrict__ out, int S, int pos0,
    int nh, int nkv, int hd, int max_seq, int qkv_stride, int out_stride) {
  const int h = blockIdx.y;
  const int t = threadIdx.x, wid = t / WAVE, lane = t % WAVE;
  const int sidx = blockIdx.x * 4 + wid;
  if (sidx >= S) return;
  const int kvh = h / (nh / nkv);
  const int e0 = 2 * lane;
  const bool act = e0 + 1 < hd;
  const int n = pos0 + sidx + 1;      // causal: attend to <= own position
  float q0 = 0.f, q1 = 0.f;
  if (act) {
    const u16* qr = qkv + (size_t)sidx * qkv_stride + (size_t)h * hd + e0;
    q0 = b2f(qr[0]);
    q1 = b2f(qr[1]);
  }
  const float scale = rsqrtf((float)hd);
  float m = -INFINITY, l = 0.f, o0 = 0.f, o1 = 0.f;
  const u16* kbase = kc + (size_t)kvh * max_seq * hd;
  const u16* vbase = vc + (size_t)kvh * max_seq * hd;
  for (int p = 0; p < n; ++p) {
    float dot = 0.f;
    if (act) {
      const u16* kr = kbase + (size_t)p * hd + e0;
      dot = q0 * b2f(kr[0]) + q1 * b2f(kr[1]);
    }
    dot = wave_sum(dot) * scale;
    dot = __shfl(dot, 0, WAVE);
    float mn = fmaxf(m, dot);
    float alpha = __expf(m - mn);
    float pw = __expf(dot - mn);
    float v0 = 0.f, v1 = 0.f;
    if (act) {
      const u16* vr = vbase + (size_t)p * hd + e0;
      v0 = b2f(vr[0]);
      v1 = b2f(vr[1]);
    }
    o0 = o0 * alpha + pw * v0;
    o1 = o1 * alpha + pw * v1;
    l = l * alpha + pw;
    m = mn;
  }
  if (act) {
    u16* orow = out + (size_t)sidx * out_stride + (size_t)h * hd + e0;
    orow[0] = f2b(o0 / l);
    orow[1] = f2b(o1 / l);
  }
}

// standalone rope for the op-level parity surface: x (BH, S, D) bf16 in
// place, cos/sin (S, D/2) f32 (backends/mod.rs:444-482 layout)
__global__ void k_rope_simple(u16* __restrict__ x,
                              const float* __restrict__ cost,
                              const float* __restrict__ sint, int s, int d) {
  const int row = blockIdx.x;  // (bh, s) flattened
  const int si = row % s;
  const int half = d / 2;
  u16* xr = x + (size_t)row * d;
  const float* c = cost + (size_t)si * half;
  const float* sn = sint + (size_t)si * half;
  for (int i = threadIdx.x; i < half; i += blockDim.x) {
    float x1 = b2f(xr[i]), x2 = b2f(xr[i + half]);
    xr[i] = f2b(x1 * c[i] - x2 * sn[i]);
    xr[i + half] = f2b(x2 * c[i] + x1 * sn[i]);
  }
}

// ---------------------------------------------------------------------------
// MFMA flash-attention prefill (hd == 128) — causal, GQA, f32 online
// softmax, bf16 I/O.  Structure after the guide's 8-wave 32x32 ladder
// (cdna_hip_programming.md §B "Fused attention prefill"), basic variant:
//   - 8 waves per workgroup, each wave owns 32 query rows of one head
//   - swapped QK^T: P = mfma(K_tile, Q_tile) so the softmax column is
//     lane-local (col j = q = lane&31); 32x32x16 bf16 MFMA, 8 per tile
//   - P -> bf16 pack + __shfl_xor(32) half-exchange assembles the PV
//     A-fragments in-register (the T12 idea without inline asm)
//   - PV reads V from a TRANSPOSED cache (vt[kvh][d][pos]) so the
//     B-fragment's 8-consecutive-k elements are one 16-B load
// Per 32-kv tile per wave: 16 MFMAs + ~16 16-B global loads; no LDS, no
// barriers (waves fully independent).
// ---------------------------------------------------------------------------
using f32x16 = __attribute__((ext_vector_type(16))) float;

__global__ __launch_bounds__(512) void k_attn_prefill_mfma(
    const u16* __restrict__ qkv, const u16* __restrict__ kc,
    const u16* __restrict__ vtc, u16* __restrict__ out, int S, int pos0,
    int nh, int nkv, int max_seq, int qkv_stride, int out_stride) {
  const int hd = 128;
  const int w = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int lhalf = lane >> 5, lq = lane & 31;
  const int h = blockIdx.y;
  const int kvh = h / (nh / nkv);
  const int qb = blockIdx.x * 256 + w * 32;
  if (qb >= S) return;

  // Q fragments (persistent): B[k][j=q], lane holds q = lq, dims
  // kk*16 + lhalf*8 .. +8
  bf16x8 qf[8];
  {
    const int row = min(qb + lq, S - 1);
    const u16* qrow = qkv + (size_t)row * qkv_stride + (size_t)h * hd;
#pragma unroll
    for (int kk = 0; kk < 8; ++kk)
      qf[kk] = *reinterpret_cast<const bf16x8*>(qrow + kk * 16 + lhalf * 8);
  }

  f32x16 oacc[4];
#pragma unroll
  for (int db = 0; db < 4; ++db)
#pragma unroll
    for (int r = 0; r < 16; ++r) oacc[db][r] = 0.f;
  float m = -INFINITY, l = 0.f;
  const float scale = rsqrtf((float)hd);
  const int q_abs = pos0 + qb + lq;
  const bool q_valid = qb + lq < S;
  const int n_wave = pos0 + min(qb + 32, S);  // kv needed by this wave
  const int ntiles = (n_wave + 31) / 32;
  const u16* kbase = kc + (size_t)kvh * max_seq * hd;
  const u16* vtbase = vtc + (size_t)kvh * hd * max_seq;

  for (int tile = 0; tile < ntiles; ++tile) {
    const int pkv = tile * 32;
    // K fragments: A[i=kv][k], lane holds kv = lq
    bf16x8 kf[8];
    {
      const u16* krow = kbase + (size_t)(pkv + lq) * hd;
#pragma unroll
      for (int kk = 0; kk < 8; ++kk)
        kf[kk] =
            *reinterpret_cast<const bf16x8*>(krow + kk * 16 + lhalf * 8);
    }
    f32x16 p;
#pragma unroll
    for (int r = 0; r < 16; ++r) p[r] = 0.f;
#pragma unroll
    for (int kk = 0; kk < 8; ++kk)
      p = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf[kk], qf[kk], p, 0, 0, 0);
    // scale + causal mask (reg r -> kv row (r&3)+8*(r>>2)+4*lhalf)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int kv_abs = pkv + (r & 3) + 8 * (r >> 2) + 4 * lhalf;
      p[r] = (q_valid && kv_abs <= q_abs) ? p[r] * scale : -INFINITY;
    }
    // online softmax stats for column q = lq (halves combined via xor-32)
    float tm = -INFINITY;
#pragma unroll
    for (int r = 0; r < 16; ++r) tm = fmaxf(tm, p[r]);
    tm = fmaxf(tm, __shfl_xor(tm, 32, WAVE));
    const float mnew = fmaxf(m, tm);
    const float alpha = (mnew == -INFINITY) ? 0.f : __expf(m - mnew);
    float ep[16];
    float tsum = 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      ep[r] = (p[r] == -INFINITY) ? 0.f : __expf(p[r] - mnew);
      tsum += ep[r];
    }
    tsum += __shfl_xor(tsum, 32, WAVE);
    l = l * alpha + tsum;
    m = mnew;
    // pack expP to bf16 pairs and exchange halves: after this each lane can
    // assemble A[i=q=lq][k=kv] fragments for PV
    u32 pk[8], rcv[8];
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      pk[i] = (u32)f2b(ep[2 * i]) | ((u32)f2b(ep[2 * i + 1]) << 16);
      rcv[i] = __shfl_xor(pk[i], 32, WAVE);
    }
    // per-row alpha for the O rescale (row q via lane shuffle)
    float arow[16];
#pragma unroll
    for (int r = 0; r < 16; ++r)
      arow[r] = __shfl(alpha, (r & 3) + 8 * (r >> 2) + 4 * lhalf, WAVE);
#pragma unroll
    for (int db = 0; db < 4; ++db)
#pragma unroll
      for (int r = 0; r < 16; ++r) oacc[db][r] *= arow[r];
    // PV: two K=16 windows over the 32-kv tile
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      union { u32 u[4]; bf16x8 v; } af;
      if (lhalf == 0) {
        af.u[0] = pk[4 * kk];
        af.u[1] = pk[4 * kk + 1];
        af.u[2] = rcv[4 * kk];
        af.u[3] = rcv[4 * kk + 1];
      } else {
        af.u[0] = rcv[4 * kk + 2];
        af.u[1] = rcv[4 * kk + 3];
        af.u[2] = pk[4 * kk + 2];
        af.u[3] = pk[4 * kk + 3];
      }
#pragma unroll
      for (int db = 0; db < 4; ++db) {
        const int d = db * 32 + lq;
        const u16* vt =
            vtbase + (size_t)d * max_seq + pkv + 16 * kk + lhalf * 8;
        bf16x8 vf = *reinterpret_cast<const bf16x8*>(vt);
        oacc[db] =
            __builtin_amdgcn_mfma_f32_32x32x16_bf16(af.v, vf, oacc[db], 0,
                                                    0, 0);
      }
    }
  }

  // epilogue: divide by l (per q row, via shuffle) and store
  float lrow[16];
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    float lv = __shfl(l, (r & 3) + 8 * (r >> 2) + 4 * lhalf, WAVE);
    lrow[r] = 1.f / lv;
  }
#pragma unroll
  for (int db = 0; db < 4; ++db) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int qrow = (r & 3) + 8 * (r >> 2) + 4 * lhalf;
      const int srow = qb + qrow;
      if (srow < S)
        out[(size_t)srow * out_stride + (size_t)h * hd + db * 32 + lq] =
            f2b(oacc[db][r] * lrow[r]);
    }
  }
}

// ---------------------------------------------------------------------------
// argmax (greedy sampling, text_model.rs:104): first-index tie-break to
// match np.argmax / candle ArgMax.  Two passes over f32 logits.
// ---------------------------------------------------------------------------
__global__ void k_argmax_part(const float* __restrict__ logits, int n,
                              float* __restrict__ pval, int* __restrict__ pidx,
                              int nparts) {
  const int part = blockIdx.x;
  const int span = (n + nparts - 1) / nparts;
  const int start = part * span, end = min(start + span, n);
  float best = -INFINITY;
  int bidx = 0x7fffffff;
  for (int i = start + (int)threadIdx.x; i < end; i += blockDim.x) {
    float v = logits[i];
    if (v > best || (v == best && i < bidx)) { best = v; bidx = i; }
  }
  // wave+block reduce keeping first index on ties
  __shared__ float sv[256];
  __shared__ int si[256];
  sv[threadIdx.x] = best;
  si[threadIdx.x] = bidx;
  __syncthreads();
  for (int off = blockDim.x / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) {
      float ov = sv[threadIdx.x + off];
      int oi = si[threadIdx.x + off];
      if (ov > sv[threadIdx.x] ||
          (ov == sv[threadIdx.x] && oi < si[threadIdx.x])) {
        sv[threadIdx.x] = ov;
        si[threadIdx.x] = oi;
      }
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) { pval[part] = sv[0]; pidx[part] = si[0]; }
}

// Final reduce; also appends the winning token to the output ring and
// advances the device position (end-of-decode-step bookkeeping).
__global__ void k_argmax_fin(const float* __restrict__ pval,
                             const int* __restrict__ pidx, int nparts,
                             u32* __restrict__ tok, int* __restrict__ pos,
                             u32* __restrict__ ring, int* __restrict__ step,
                             int advance_pos) {
  __shared__ float sv[256];
  __shared__ int si[256];
  float best = -INFINITY;
  int bidx = 0x7fffffff;
  for (int i = threadIdx.x; i < nparts; i += blockDim.x) {
    float v = pval[i];
    if (v > best || (v == best && pidx[i] < bidx)) { best = v; bidx = pidx[i]; }
  }
  sv[threadIdx.x] = best;
  si[threadIdx.x] = bidx;
  __syncthreads();
  for (int off = blockDim.x / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) {
      float ov = sv[threadIdx.x + off];
      int oi = si[threadIdx.x + off];
      if (ov > sv[threadIdx.x] ||
          (ov == sv[threadIdx.x] && oi < si[threadIdx.x])) {
        sv[threadIdx.x] = ov;
        si[threadIdx.x] = oi;
      }
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    *tok = (u32)si[0];
    if (ring) {
      ring[*step] = (u32)si[0];
      *step += 1;
    }
    if (advance_pos) *pos += 1;
  }
}

__global__ void k_advance_pos(int* __restrict__ pos, int by) {
  if (threadIdx.x == 0) *pos += by;
}

// ---------------------------------------------------------------------------
// Prefill GEMM — MFMA bf16.  C[M,N] = A[M,K] @ W[N,K]^T (+ residual).
// 128x128 tile, BK=64, 256 threads (4 waves as 2x2 of 64x64 sub-tiles),
// v_mfma_f32_16x16x32_bf16, double-buffered LDS filled by
// global_load_lds_dwordx4 with the XOR source-swizzle (rule 21) so the
// fragment ds_read_b128 is bank-conflict-free (guide §5 ladder step 3).
// Row clamp handles M/N tails; requires K % 64 == 0 (all hot-path K are).
// ---------------------------------------------------------------------------
#define GEMM_BM 128
#define GEMM_BN 128
#define GEMM_BK 64

template <int EPI>  // 0: bf16 store, 1: bf16 store + residual add
__global__ __launch_bounds__(256) void k_gemm_bf16(
    const u16* __restrict__ A, const u16* __restrict__ W, u16* __restrict__ C,
    const u16* __restrict__ res, int M, int N, int K) {
  // dynamic-LDS base must be 16-B aligned (guide G17)
  extern __shared__ __attribute__((aligned(16))) char smem[];
  u16* As = reinterpret_cast<u16*>(smem);                       // [2][128][64]
  u16* Bs = reinterpret_cast<u16*>(smem + 2 * GEMM_BM * GEMM_BK * 2);

  // XCD-aware block swizzle (guide T1, bijective variant): contiguous
  // chunks of the grid per XCD so neighbor tiles hit the same per-XCD L2
  int mt, nt;
  {
    const int nwg = gridDim.x * gridDim.y;
    const int orig = blockIdx.x + gridDim.x * blockIdx.y;
    const int q = nwg / 8, rr = nwg % 8;
    const int xcd = orig % 8, idx = orig / 8;
    const int wgid =
        (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
    mt = wgid % gridDim.x;
    nt = wgid / gridDim.x;
  }
  const int m0 = mt * GEMM_BM;
  const int n0 = nt * GEMM_BN;
  const int t = threadIdx.x;
  const int wid = t / WAVE, lane = t % WAVE;
  const int wr = wid / 2, wc = wid % 2;  // wave's 64x64 quadrant

  // staging geometry: each wave issues 4 glds per 16 KB tile; instruction j
  // of wave w writes LDS rows [(w*4+j)*8, +8) (8 rows x 128 B), lane l ->
  // row sub = l/8, 16-B unit u = l%8, with source column unit u ^ (row & 7).
  auto stage = [&](int buf, int kt) {
    const int ktbase = kt * GEMM_BK;
    u16* as = As + (size_t)buf * GEMM_BM * GEMM_BK;
    u16* bs = Bs + (size_t)buf * GEMM_BN * GEMM_BK;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int r = (wid * 4 + j) * 8 + lane / 8;
      const int u = (lane % 8) ^ (r & 7);
      {  // A tile
        int grow = min(m0 + r, M - 1);
        const u16* src = A + (size_t)grow * K + ktbase + u * 8;
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) void*)(uintptr_t)src,
            (__attribute__((address_space(3))) void*)(uintptr_t)(
                as + (size_t)(wid * 4 + j) * 8 * GEMM_BK),
            16, 0, 0);
      }
      {  // W tile (B^T)
        int grow = min(n0 + r, N - 1);
        const u16* src = W + (size_t)grow * K + ktbase + u * 8;
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) void*)(uintptr_t)src,
            (__attribute__((address_space(3))) void*)(uintptr_t)(
                bs + (size_t)(wid * 4 + j) * 8 * GEMM_BK),
            16, 0, 0);
      }
    }
  };

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int ntiles = K / GEMM_BK;
  stage(0, 0);
  __syncthreads();  // drains the glds (vmcnt(0) implied by the barrier)

  for (int kt = 0; kt < ntiles; ++kt) {
    const int cur = kt & 1;
    if (kt + 1 < ntiles) stage(cur ^ 1, kt + 1);
    const u16* as = As + (size_t)cur * GEMM_BM * GEMM_BK;
    const u16* bs = Bs + (size_t)cur * GEMM_BN * GEMM_BK;
#pragma unroll
    for (int ks = 0; ks < GEMM_BK / 32; ++ks) {
      bf16x8 af[4], bf[4];
      const int kk = ks * 32 + (lane / 16) * 8;
      const int ku = kk / 8;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int ar = wr * 64 + i * 16 + (lane % 16);
        af[i] = *reinterpret_cast<const bf16x8*>(
            as + (size_t)ar * GEMM_BK + (size_t)(ku ^ (ar & 7)) * 8);
        const int br = wc * 64 + i * 16 + (lane % 16);
        bf[i] = *reinterpret_cast<const bf16x8*>(
            bs + (size_t)br * GEMM_BK + (size_t)(ku ^ (br & 7)) * 8);
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[i], bf[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  // epilogue: C/D map (guide §3): col = lane&15, row = (lane>>4)*4 + reg
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + wr * 64 + i * 16 + (lane / 16) * 4 + r;
        const int col = n0 + wc * 64 + j * 16 + (lane % 16);
        if (row < M && col < N) {
          float v = acc[i][j][r];
          if (EPI == 1) v += b2f(res[(size_t)row * N + col]);
          C[(size_t)row * N + col] = f2b(v);
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// 256x256 GEMM with counted-vmcnt pipelining (guide §5 T3/T4: the counted
// wait IS the gain — never drain vmcnt(0) in the main loop).
//
// Geometry: BM=BN=256, BK=64, 8 waves (2M x 4N), per-wave 128x64 output
// (8x4 16x16 frags, 128 acc VGPRs).  LDS 128 KB: 2 tile-buffers x
// {A,B} x 2 row-halves x [128][64] bf16, my known-good XOR-unit swizzle.
//
// Schedule (uniform, every phase): stage 2 row-half pieces of tile t+1
// (4 glds/wave) -> asm s_waitcnt vmcnt(4) (retires everything except the 4
// just-issued, so tile t is fully resident) -> raw s_barrier -> 12
// ds_read_b128 + 32 MFMA (one k-step) -> next phase.  Piece issue order per
// wave: (t,A-h0)(t,A-h1) in phase ks=0, (t,B-h0)(t,B-h1) in ks=1, so the
// FIFO count proof holds; buffers alternate by tile parity and a piece is
// overwritten two tiles after its last read (>= 2 barriers apart).
// Requires K % 64 == 0; M/N tails handled by clamped loads + guarded
// stores.  Dispatched only when the grid has >= GEMM256_MIN_BLOCKS tiles.
// ---------------------------------------------------------------------------
template <int EPI>
__global__ __launch_bounds__(512) void k_gemm_256(
    const u16* __restrict__ A, const u16* __restrict__ W, u16* __restrict__ C,
    const u16* __restrict__ res, int M, int N, int K) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // piece (op, buf, half): As op=0 at 0, Bs op=1 at 64 KB
  auto piece = [&](int op, int buf, int half) -> u16* {
    return reinterpret_cast<u16*>(smem + op * 65536 +
                                  (buf * 2 + half) * 16384);
  };
  // XCD-aware bijective block swizzle (T1)
  int mt, nt;
  {
    const int nwg = gridDim.x * gridDim.y;
    const int orig = blockIdx.x + gridDim.x * blockIdx.y;
    const int q = nwg / 8, rr = nwg % 8;
    const int xcd = orig % 8, idx = orig / 8;
    const int wgid =
        (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
    mt = wgid % gridDim.x;
    nt = wgid / gridDim.x;
  }
  const int m0 = mt * 256, n0 = nt * 256;
  const int t = threadIdx.x;
  const int w = t / WAVE, lane = t % WAVE;
  const int wm = w >> 2, wn = w & 3;

  // stage the two row-half pieces of ONE operand for tile tau
  auto stage_op = [&](int op, int tau) {
    const u16* G = op == 0 ? A : W;
    const int rows = op == 0 ? M : N;
    const int base = op == 0 ? m0 : n0;
    const int buf = tau & 1;
    const int kt = tau * 64;
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      u16* dst = piece(op, buf, half);
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        const int rloc = (w * 2 + j) * 8 + lane / 8;   // 0..127
        const int grow = min(base + half * 128 + rloc, rows - 1);
        const int u = (lane % 8) ^ (rloc & 7);
        const u16* src = G + (size_t)grow * K + kt + u * 8;
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) void*)(uintptr_t)src,
            (__attribute__((address_space(3))) void*)(uintptr_t)(
                dst + (size_t)(w * 2 + j) * 8 * 64),
            16, 0, 0);
      }
    }
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int NT = K / 64;
  // prologue: tile 0, both operands (8 glds/wave in flight)
  stage_op(0, 0);
  stage_op(1, 0);
  for (int tau = 0; tau < NT; ++tau) {
    const int buf = tau & 1;
    const u16* ap = piece(0, buf, wm);               // this wave's A half
    const u16* bp = piece(1, buf, wn >> 1);          // this wave's B half
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      if (tau + 1 < NT) {
        stage_op(ks, tau + 1);                       // ks0: A, ks1: B
        // tile tau fully resident once all but the 4 just-issued retire
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
      } else {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // last tile
      }
      __builtin_amdgcn_s_barrier();
      bf16x8 af[8], bf[4];
      const int u = ks * 4 + (lane / 16);            // 16-B unit 0..7
#pragma unroll
      for (int mi = 0; mi < 8; ++mi) {
        const int r = mi * 16 + (lane % 16);
        af[mi] = *reinterpret_cast<const bf16x8*>(
            ap + (size_t)r * 64 + (size_t)(u ^ (r & 7)) * 8);
      }
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const int r = (wn & 1) * 64 + ni * 16 + (lane % 16);
        bf[ni] = *reinterpret_cast<const bf16x8*>(
            bp + (size_t)r * 64 + (size_t)(u ^ (r & 7)) * 8);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mi = 0; mi < 8; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();
    }
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // quiesce before exit

  // epilogue (C/D map: col = lane&15, row = (lane>>4)*4 + reg)
#pragma unroll
  for (int mi = 0; mi < 8; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + wm * 128 + mi * 16 + (lane / 16) * 4 + r;
        const int col = n0 + wn * 64 + ni * 16 + (lane % 16);
        if (row < M && col < N) {
          float v = acc[mi][ni][r];
          if (EPI == 1) v += b2f(res[(size_t)row * N + col]);
          C[(size_t)row * N + col] = f2b(v);
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// launch wrappers (host side) — declared in kernels.h, used by engine.hip
// ---------------------------------------------------------------------------
#include "kernels.h"

void launch_f32_to_bf16(const float* in, u16* out, size_t n, hipStream_t s) {
  int blocks = (int)min((n + 255) / 256, (size_t)2048);
  hipLaunchKernelGGL(k_f32_to_bf16, dim3(blocks), dim3(256), 0, s, in, out, n);
}
void launch_bf16_to_f32(const u16* in, float* out, size_t n, hipStream_t s) {
  int blocks = (int)min((n + 255) / 256, (size_t)2048);
  hipLaunchKernelGGL(k_bf16_to_f32, dim3(blocks), dim3(256), 0, s, in, out, n);
}
void launch_fill_random(u16* out, size_t n, uint64_t seed, float scale,
                        hipStream_t s) {
  hipLaunchKernelGGL(k_fill_random, dim3(2048), dim3(256), 0, s, out, n, seed,
                     scale);
}
void launch_fill_random_u8(unsigned char* out, size_t n, uint64_t seed,
                           hipStream_t s) {
  hipLaunchKernelGGL(k_fill_random_u8, dim3(2048), dim3(256), 0, s, out, n,
                     seed);
}
void launch_fill_const(u16* out, size_t n, float v, hipStream_t s) {
  hipLaunchKernelGGL(k_fill_const, dim3(512), dim3(256), 0, s, out, n, v);
}
void launch_rmsnorm(const u16* x, const u16* w, u16* out, int rows, int cols,
                    float eps, hipStream_t s) {
  hipLaunchKernelGGL(k_rmsnorm, dim3(rows), dim3(256), 0, s, x, w, out, cols,
                     eps, rows, (size_t)0);
}
void launch_rmsnorm_strided(const u16* x, const u16* w, u16* out, int outer,
                            int inner, size_t outer_stride, int cols,
                            float eps, hipStream_t s) {
  hipLaunchKernelGGL(k_rmsnorm, dim3(outer * inner), dim3(256), 0, s, x, w,
                     out, cols, eps, inner, outer_stride);
}
void launch_silu_mul_rows(const u16* gu, u16* out, int S, int I,
                          hipStream_t s) {
  size_t n = (size_t)S * I;
  int blocks = (int)min((n + 255) / 256, (size_t)4096);
  hipLaunchKernelGGL(k_silu_mul_rows, dim3(blocks), dim3(256), 0, s, gu, out,
                     S, I);
}
void launch_silu_mul(const u16* g, const u16* u, u16* out, size_t n,
                     hipStream_t s) {
  int blocks = (int)min((n + 255) / 256, (size_t)4096);
  hipLaunchKernelGGL(k_silu_mul, dim3(blocks), dim3(256), 0, s, g, u, out, n);
}
template <int ROWS, int EPI>
static void gemv_dispatch_kb(const u16* W, const u16* x, void* out,
                             const u16* res, const u16* nw, float eps, int N,
                             int K, hipStream_t s) {
  dim3 grid((N + ROWS - 1) / ROWS);
  if (K > 16384) {
    hipLaunchKernelGGL((k_gemv_stream<ROWS, EPI>), grid, dim3(256), 0, s, W,
                       x, out, res, N, K);
    return;
  }
#define GEMV_KB(KB)                                                        \
  do {                                                                     \
    if (nw)                                                                \
      hipLaunchKernelGGL((k_gemv_reg<ROWS, EPI, true, KB>), grid,          \
                         dim3(256), 0, s, W, x, out, res, nw, eps, N, K);  \
    else                                                                   \
      hipLaunchKernelGGL((k_gemv_reg<ROWS, EPI, false, KB>), grid,         \
                         dim3(256), 0, s, W, x, out, res, nw, eps, N, K);  \
  } while (0)
  if (K <= 4096) GEMV_KB(2);
  else if (K <= 8192) GEMV_KB(4);
  else GEMV_KB(8);
#undef GEMV_KB
}

void launch_gemv(const u16* W, const u16* x, void* out, const u16* res,
                 const u16* nw, float eps, int N, int K, int epi,
                 hipStream_t s) {
  // small row-count per block keeps the grid >= ~2048 workgroups on the
  // usual decode shapes (256 CUs need many blocks to reach HBM peak)
  static const int env_small = [] {
    const char* v = getenv("CAKE_GEMV_ROWS_SMALL");
    return v ? atoi(v) : 0;
  }();
  int rows;
  if (N >= 16384) rows = 8;                 // lm_head: maximal block count
  else if (env_small) rows = env_small;     // A/B override
  else if (K >= 8192) rows = 4;             // long rows amortize x reload
  else rows = (N >= 6144) ? 4 : 2;
#define GEMV_R(R, EPI) gemv_dispatch_kb<R, EPI>(W, x, out, res, nw, eps, N, K, s)
#define GEMV_EPI(EPI)                          \
  do {                                         \
    if (rows >= 8) GEMV_R(8, EPI);             \
    else if (rows == 4) GEMV_R(4, EPI);        \
    else if (rows == 2) GEMV_R(2, EPI);        \
    else GEMV_R(1, EPI);                       \
  } while (0)
  if (epi == 0) GEMV_EPI(0);
  else if (epi == 1) GEMV_EPI(1);
  else GEMV_EPI(2);
#undef GEMV_EPI
#undef GEMV_R
}
void launch_gemv_gateup(const u16* W, const u16* x, u16* out, const u16* nw,
                        float eps, int I, int K, int rows, hipStream_t s) {
  if (K > 16384) return;  // guarded at engine create (K = hidden <= 16384)
#define GU_KB2(ROWS, KB)                                                    \
  do {                                                                      \
    dim3 grid((I + ROWS - 1) / ROWS);                                       \
    if (nw)                                                                 \
      hipLaunchKernelGGL((k_gemv_gateup<ROWS, true, KB>), grid, dim3(256),  \
                         0, s, W, x, out, nw, eps, I, K);                   \
    else                                                                    \
      hipLaunchKernelGGL((k_gemv_gateup<ROWS, false, KB>), grid, dim3(256), \
                         0, s, W, x, out, nw, eps, I, K);                   \
  } while (0)
#define GU_KB(KB)                                                           \
  do {                                                                      \
    if (rows >= 8) GU_KB2(8, KB);                                           \
    else GU_KB2(4, KB);                                                     \
  } while (0)
  if (K <= 4096) GU_KB(2);
  else if (K <= 8192) GU_KB(4);
  else GU_KB(8);
#undef GU_KB
#undef GU_KB2
}
void launch_gemv_fp8(const unsigned char* W, const float* sc, const u16* x,
                     void* out, const u16* res, const u16* nw, float eps,
                     int N, int K, int epi, hipStream_t s) {
  const int nkb = (K + 127) / 128;
  static const int env_rows = [] {
    const char* v = getenv("CAKE_FP8_ROWS");
    return v ? atoi(v) : 0;
  }();
  const int rows = N >= 16384 ? 8 : (env_rows ? env_rows : 4);
#define F8_KB(R, EPI, KB)                                                   \
  do {                                                                      \
    dim3 grid((N + R - 1) / R);                                             \
    if (nw)                                                                 \
      hipLaunchKernelGGL((k_gemv_fp8<R, EPI, true, KB>), grid, dim3(256),   \
                         0, s, W, sc, x, out, res, nw, eps, N, K, nkb);     \
    else                                                                    \
      hipLaunchKernelGGL((k_gemv_fp8<R, EPI, false, KB>), grid, dim3(256),  \
                         0, s, W, sc, x, out, res, nw, eps, N, K, nkb);     \
  } while (0)
#define F8_R(R, EPI)                                                        \
  do {                                                                      \
    if (K > 16384) {                                                        \
      dim3 grid((N + R - 1) / R);                                           \
      hipLaunchKernelGGL((k_gemv_fp8_stream<R, EPI>), grid, dim3(256), 0,   \
                         s, W, sc, x, out, res, N, K, nkb);                 \
    } else if (K <= 4096) F8_KB(R, EPI, 1);                                 \
    else if (K <= 8192) F8_KB(R, EPI, 2);                                   \
    else F8_KB(R, EPI, 4);                                                  \
  } while (0)
  if (epi == 0) { if (rows == 8) F8_R(8, 0); else F8_R(4, 0); }
  else if (epi == 1) { if (rows == 8) F8_R(8, 1); else F8_R(4, 1); }
  else { if (rows == 8) F8_R(8, 2); else F8_R(4, 2); }
#undef F8_R
#undef F8_KB
}
void launch_gemv_gateup_fp8(const unsigned char* W, const float* sc,
                            const u16* x, u16* out, const u16* nw, float eps,
                            int I, int K, hipStream_t s) {
  const int nkb = (K + 127) / 128;
  dim3 grid((I + 3) / 4);
#define GU8_KB(KB)                                                          \
  do {                                                                      \
    if (nw)                                                                 \
      hipLaunchKernelGGL((k_gemv_gateup_fp8<4, true, KB>), grid, dim3(256), \
                         0, s, W, sc, x, out, nw, eps, I, K, nkb);          \
    else                                                                    \
      hipLaunchKernelGGL((k_gemv_gateup_fp8<4, false, KB>), grid,           \
                         dim3(256), 0, s, W, sc, x, out, nw, eps, I, K,     \
                         nkb);                                              \
  } while (0)
  if (K <= 4096) GU8_KB(1);
  else if (K <= 8192) GU8_KB(2);
  else GU8_KB(4);
#undef GU8_KB
}
void launch_dequant_fp8(const unsigned char* W, const float* sc, u16* out,
                        int N, int K, hipStream_t s) {
  const int nkb = (K + 127) / 128;
  hipLaunchKernelGGL(k_dequant_fp8, dim3(4096), dim3(256), 0, s, W, sc, out,
                     N, K, nkb);
}
void launch_embed_token(const u16* embed, const u32* tok, u16* x, int H,
                        hipStream_t s) {
  hipLaunchKernelGGL(k_embed_token, dim3(1), dim3(256), 0, s, embed, tok, x, H);
}
void launch_embed_rows(const u16* embed, const u32* ids, u16* x, int S, int H,
                       hipStream_t s) {
  hipLaunchKernelGGL(k_embed_rows, dim3(S), dim3(256), 0, s, embed, ids, x, H);
}
void launch_rope_store_decode(u16* qkv, u16* kc, u16* vc, u16* vtc,
                              const float* cost, const float* sint,
                              const int* pos, int nh, int nkv, int hd, int rd,
                              int max_seq, const u16* qn, const u16* kn,
                              float eps, hipStream_t s) {
  hipLaunchKernelGGL(k_rope_store_decode, dim3(nh + 2 * nkv), dim3(64), 0, s,
                     qkv, kc, vc, vtc, cost, sint, pos, nh, nkv, hd, rd,
                     max_seq, qn, kn, eps);
}
void launch_rope_store_prefill(u16* qkv, u16* kc, u16* vc, u16* vtc,
                               const float* cost, const float* sint, int pos0,
                               int S, int nh, int nkv, int hd, int rd,
                               int max_seq, int qkv_stride, const u16* qn,
                               const u16* kn, float eps, hipStream_t s) {
  hipLaunchKernelGGL(k_rope_store_prefill, dim3(nh + 2 * nkv, S), dim3(64), 0,
                     s, qkv, kc, vc, vtc, cost, sint, pos0, nh, nkv, hd, rd,
                     max_seq, qkv_stride, qn, kn, eps);
}
void launch_attn_decode(const u16* q, const u16* kc, const u16* vc,
                        const int* pos, float* ws, u32* cnt, u16* out, int nh,
                        int nkv, int hd, int max_seq, int nchunk,
                        hipStream_t s) {
  hipLaunchKernelGGL(k_attn_decode_fused, dim3(nchunk, nh), dim3(256), 0, s,
                     q, kc, vc, pos, ws, cnt, out, nh, nkv, hd, max_seq,
                     nchunk);
}
void launch_attn_prefill(const u16* qkv, const u16* kc, const u16* vc,
                         const u16* vtc, u16* out, int S, int pos0, int nh,
                         int nkv, int hd, int max_seq, int qkv_stride,
                         int out_stride, hipStream_t s) {
  if (hd == 128) {
    hipLaunchKernelGGL(k_attn_prefill_mfma, dim3((S + 255) / 256, nh),
                       dim3(512), 0, s, qkv, kc, vtc, out, S, pos0, nh, nkv,
                       max_seq, qkv_stride, out_stride);
  } else {
    hipLaunchKernelGGL(k_attn_prefill, dim3((S + 3) / 4, nh), dim3(256), 0, s,
                       qkv, kc, vc, out, S, pos0, nh, nkv, hd, max_seq,
                       qkv_stride, out_stride);
  }
}
void launch_rope_simple(u16* x, const float* cost, const float* sint, int bh,
                        int s, int d, hipStream_t st) {
  hipLaunchKernelGGL(k_rope_simple, dim3(bh * s), dim3(64), 0, st, x, cost,
                     sint, s, d);
}
void launch_argmax(const float* logits, int n, float* pval, int* pidx,
                   u32* tok, int* pos, u32* ring, int* step, int advance_pos,
                   hipStream_t s) {
  const int nparts = 256;
  hipLaunchKernelGGL(k_argmax_part, dim3(nparts), dim3(256), 0, s, logits, n,
                     pval, pidx, nparts);
  hipLaunchKernelGGL(k_argmax_fin, dim3(1), dim3(256), 0, s, pval, pidx,
                     nparts, tok, pos, ring, step, advance_pos);
}
void launch_advance_pos(int* pos, int by, hipStream_t s) {
  hipLaunchKernelGGL(k_advance_pos, dim3(1), dim3(64), 0, s, pos, by);
}
void launch_gemm(const u16* A, const u16* W, u16* C, const u16* res, int M,
                 int N, int K, int epi, hipStream_t s) {
  // 256^2 counted-vmcnt kernel when the grid still fills the chip at one
  // 8-wave block per CU (CAKE_GEMM256=0 disables for A/B)
  static const int use256 = [] {
    const char* v = getenv("CAKE_GEMM256");
    return v ? atoi(v) : 1;
  }();
  const int mt = (M + 255) / 256, nt = (N + 255) / 256;
  if (use256 && (long)mt * nt >= 200) {
    static bool attr_set = false;
    if (!attr_set) {
      hipFuncSetAttribute((const void*)&k_gemm_256<0>,
                          hipFuncAttributeMaxDynamicSharedMemorySize, 131072);
      hipFuncSetAttribute((const void*)&k_gemm_256<1>,
                          hipFuncAttributeMaxDynamicSharedMemorySize, 131072);
      attr_set = true;
    }
    dim3 grid(mt, nt);
    if (epi == 0)
      hipLaunchKernelGGL(k_gemm_256<0>, grid, dim3(512), 131072, s, A, W, C,
                         res, M, N, K);
    else
      hipLaunchKernelGGL(k_gemm_256<1>, grid, dim3(512), 131072, s, A, W, C,
                         res, M, N, K);
    return;
  }
  dim3 grid((M + GEMM_BM - 1) / GEMM_BM, (N + GEMM_BN - 1) / GEMM_BN);
  size_t lds = 2 * (size_t)(GEMM_BM + GEMM_BN) * GEMM_BK * 2;
  if (epi == 0)
    hipLaunchKernelGGL(k_gemm_bf16<0>, grid, dim3(256), lds, s, A, W, C, res,
                       M, N, K);
  else
    hipLaunchKernelGGL(k_gemm_bf16<1>, grid, dim3(256), lds, s, A, W, C, res,
                       M, N, K);
}
