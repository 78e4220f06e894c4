// Shared device helpers for the gfx950 kernels (split into per-family
// translation units so template-instantiation churn in one family cannot
// perturb another's codegen — guide §5.4 rule 19).
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>
#include <cstdlib>

#define WAVE 64

using u16 = unsigned short;
using u32 = unsigned int;
using f32x4 = __attribute__((ext_vector_type(4))) float;
using f32x16 = __attribute__((ext_vector_type(16))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using short8 = __attribute__((ext_vector_type(8))) short;
using uint4v = __attribute__((ext_vector_type(4))) unsigned int;
using f32x2 = __attribute__((ext_vector_type(2))) float;

// bf16 <-> f32 (round-to-nearest-even, matches torch/HF casting)
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

// nontemporal 16-B weight load: decode weights are streamed exactly once
// per token — keep them from thrashing L2/L3 (guide G14)
__device__ inline short8 ntload8(const u16* p) {
  uint4v v = __builtin_nontemporal_load(reinterpret_cast<const uint4v*>(p));
  union { uint4v u; short8 s; } c{v};
  return c.s;
}
__device__ inline uint4v ntload16b(const unsigned char* p) {
  return __builtin_nontemporal_load(reinterpret_cast<const uint4v*>(p));
}
// 16 fp8 -> 16 f32 via the hardware v_cvt_pk_f32_fp8 (2 elems/instruction)
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
