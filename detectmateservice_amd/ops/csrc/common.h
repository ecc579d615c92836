// Common helpers for detectmate-mi355x HIP kernels (gfx950 / CDNA4 only).
//
// Design references: /opt/skills/guides/cdna_hip_programming.md (wave64,
// MFMA intrinsics, LDS banking, glds staging) — written for MI355X from
// scratch; no CUDA compatibility paths.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

#define DMX_WAVE 64

#define DMX_HIP_CHECK(expr)                                                    \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) {                                                    \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(_e), __FILE__,       \
             __LINE__);                                                        \
      abort();                                                                 \
    }                                                                          \
  } while (0)

// Vector types for wide loads (Guideline 13: always vectorize bf16 as
// short4/short8 reinterpret — 8-16 B/lane).
typedef __attribute__((ext_vector_type(4))) short short4v;
typedef __attribute__((ext_vector_type(8))) short short8v;
typedef __attribute__((ext_vector_type(2))) float float2v;
typedef __attribute__((ext_vector_type(4))) float float4v;
typedef __attribute__((ext_vector_type(4))) unsigned int uint4v;

// MFMA fragment types for mfma_f32_16x16x32_bf16 (guide §3: 8 bf16 input
// elements per lane = 4 VGPRs; 4 fp32 accumulator elements per lane).
typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

static __device__ __forceinline__ float bf16_to_f32(short u) {
  union { float f; unsigned int i; } v;
  v.i = ((unsigned int)(unsigned short)u) << 16;
  return v.f;
}

static __device__ __forceinline__ short f32_to_bf16(float f) {
  // lowers to v_cvt_pk_bf16_f32 (1 instr; the manual RNE bit-math version
  // cost ~5 VALU each and epilogues convert ~150 values/lane/layer)
  __hip_bfloat16 h = __float2bfloat16(f);
  union { __hip_bfloat16 h; short s; } v;
  v.h = h;
  return v.s;
}

// GELU, tanh approximation (torch gelu(approximate="tanh"); the GPT-2 /
// BERT-family standard approximation). The erf form cost ~2x the VALU
// ops (libdevice erff is a long polynomial) and the fused kernel is
// VALU-issue-bound; the framework uses the tanh form consistently
// (kernels, CPU fallbacks, tests).
static __device__ __forceinline__ float gelu_f32(float x) {
  // EXACT identity: 0.5*(1+tanh(a)) == sigmoid(2a), so
  //   gelu_tanh(x) = x * sigmoid(2c*(x + 0.044715 x^3)),
  // computable with ONE v_exp + one reciprocal instead of libm tanhf
  // (branchy, multiple transcendentals — it made the FFN phase half the
  // fused kernel's per-wave work, tools/probe_bert_timing.py).
  const float two_c = 1.5957691216057308f;  // 2*sqrt(2/pi)
  const float z = two_c * (x + 0.044715f * x * x * x);
  return x / (1.0f + __expf(-z));
}

static __device__ __forceinline__ float warp_reduce_sum_f32(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;
}

static __device__ __forceinline__ float warp_reduce_max_f32(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, 64));
  return v;
}
