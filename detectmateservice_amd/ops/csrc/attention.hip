// Fused multi-head self-attention for short sequences (log-line tokens).
//
// Shapes: Q,K,V [B, H, S, Dh] bf16 with S <= 128, Dh <= 64 (BERT-tiny
// detector: S=64, Dh=64). One workgroup per (b, h): K and V staged in LDS
// once (S*Dh*2 B each, 16 KiB at 64x64), 4 waves each own S/4 query rows.
// Per q-row the owning wave computes scores lane-parallel (lane j = key j),
// softmax via wave shuffle reduction, then O[q,:] accumulated lane-parallel
// over Dh with P broadcast by shfl — no S x S score matrix ever
// materialized (flash-style online structure is unnecessary at S<=128:
// the whole row of scores fits in registers, one per lane).
//
// VALU-dot v1 (attention is ~15% of detector FLOPs at BERT-tiny shapes;
// the linears carry the MFMA load). K staged row-major; scores read K rows
// per-lane from LDS as bf16x8 vectors.
#include "common.h"

#define ATTN_WAVES 4

extern "C" __global__ __launch_bounds__(ATTN_WAVES * DMX_WAVE)
void dmx_attention_bf16(
    const short* __restrict__ Q,  // [B*H, S, Dh]
    const short* __restrict__ K,
    const short* __restrict__ V,
    short* __restrict__ O,        // [B*H, S, Dh]
    int BH, int S, int Dh, float scale) {
  const int bh = blockIdx.x;
  if (bh >= BH) return;
  const int wid = threadIdx.x / DMX_WAVE;
  const int lane = threadIdx.x % DMX_WAVE;
  const long base = (long)bh * S * Dh;

  extern __shared__ __attribute__((aligned(16))) short smem[];
  short* k_lds = smem;          // [S][Dh]
  short* v_lds = smem + S * Dh; // [S][Dh]

  // cooperative stage: 256 threads, vector 8 (S*Dh multiple of 8*?; host
  // asserts (S*Dh) % (8*ATTN_WAVES*DMX_WAVE) == 0 or falls back to scalar)
  const int total = S * Dh;
  for (int i = threadIdx.x * 8; i < total; i += ATTN_WAVES * DMX_WAVE * 8) {
    *(short8v*)(k_lds + i) = *(const short8v*)(K + base + i);
    *(short8v*)(v_lds + i) = *(const short8v*)(V + base + i);
  }
  __syncthreads();

  const int rows_per_wave = (S + ATTN_WAVES - 1) / ATTN_WAVES;
  const int q0 = wid * rows_per_wave;
  float p_scores; // this lane's score for key j=lane (S <= 64) or two keys

  for (int qi = q0; qi < min(q0 + rows_per_wave, S); ++qi) {
    // load q row into registers (all lanes hold the full row via LDS-free
    // global read: Dh<=64, each lane reads bf16x8 chunks it needs)
    // score for key j = lane (and lane+64 if S > 64)
    float s0 = 0.f, s1 = 0.f;
    const short* qrow = Q + base + (long)qi * Dh;
#pragma unroll 4
    for (int d = 0; d < Dh; d += 8) {
      short8v qv = *(const short8v*)(qrow + d);
      short8v kv0;
      if (lane < S) kv0 = *(const short8v*)(k_lds + lane * Dh + d);
      short8v kv1;
      if (S > 64 && lane + 64 < S)
        kv1 = *(const short8v*)(k_lds + (lane + 64) * Dh + d);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float qf = bf16_to_f32(qv[j]);
        if (lane < S) s0 += qf * bf16_to_f32(kv0[j]);
        if (S > 64 && lane + 64 < S) s1 += qf * bf16_to_f32(kv1[j]);
      }
    }
    s0 = lane < S ? s0 * scale : -1e30f;
    s1 = (S > 64 && lane + 64 < S) ? s1 * scale : -1e30f;

    // softmax over up to 128 scores held 2-per-lane
    float m = fmaxf(s0, s1);
    m = warp_reduce_max_f32(m);
    m = __shfl(m, 0, 64);
    float e0 = lane < S ? __expf(s0 - m) : 0.f;
    float e1 = (S > 64 && lane + 64 < S) ? __expf(s1 - m) : 0.f;
    float denom = warp_reduce_sum_f32(e0 + e1);
    denom = __shfl(denom, 0, 64);
    const float inv = 1.f / denom;
    e0 *= inv;
    e1 *= inv;

    // O[qi, d=lane (d < Dh)]: accumulate over keys with p broadcast
    float acc = 0.f;
    for (int j = 0; j < S; ++j) {
      const float p = j < 64 ? __shfl(e0, j, 64) : __shfl(e1, j - 64, 64);
      if (lane < Dh) acc += p * bf16_to_f32(v_lds[j * Dh + lane]);
    }
    if (lane < Dh) O[base + (long)qi * Dh + lane] = f32_to_bf16(acc);
  }
}

extern "C" void dmx_launch_attention_bf16(
    const void* Q, const void* K, const void* V, void* O,
    int BH, int S, int Dh, float scale, hipStream_t stream) {
  const size_t lds = (size_t)2 * S * Dh * sizeof(short);
  hipLaunchKernelGGL(dmx_attention_bf16, dim3(BH), dim3(ATTN_WAVES * DMX_WAVE),
                     lds, stream, (const short*)Q, (const short*)K,
                     (const short*)V, (short*)O, BH, S, Dh, scale);
}
