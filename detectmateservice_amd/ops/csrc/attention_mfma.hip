// MFMA fused multi-head self-attention for short sequences (gfx950).
//
// Replaces the VALU path of attention.hip for the hot shapes (S%32==0,
// Dh in {32,64}): one workgroup per (batch, head), 4 waves, each wave owns
// 16 query rows. Reads Q/K/V straight from the fused QKV projection
// output [B, S, 3*H*Dh] (no permute/contiguous copies — those were ~6% of
// step time in profiles/r01_bench_b32768_kernel_stats.txt) and writes O in
// [B, S, H*Dh] so the output projection consumes it directly.
//
// Per (b,h):
//   LDS:  K  [S][Dh+8]   k-contiguous, +8 bf16 pad = 16-B-aligned rows,
//                        row-stride 144 B -> bank-spread ds_read_b128
//         Vt [Dh][S+8]   V transposed at staging so the PV B-operand is
//                        j-contiguous (same fragment pattern as K)
//         P  [4][16][S+8] per-wave probability tile (bf16)
//   QK^T: mfma_f32_16x16x32_bf16, A-frags streamed from global Q,
//         B-frags from K LDS; full-row softmax in the accumulator layout
//         (row = (lane>>4)*4+r, col = lane&15) via 16-lane shfl_xor
//         reductions; P written unnormalized to LDS, O scaled by 1/rowsum
//         in the epilogue.
//   PV:   A-frags from P LDS, B-frags from Vt LDS.
#include "common.h"

#define AM_WAVES 4

extern "C" __global__ __launch_bounds__(AM_WAVES * DMX_WAVE)
void dmx_attention_mfma_bf16(
    const short* __restrict__ QKV,  // [B, S, 3*H*Dh]
    short* __restrict__ O,          // [B, S, H*Dh]
    int Batch, int S, int H, int Dh, float scale) {
  const int bh = blockIdx.x;
  const int b = bh / H;
  const int h = bh % H;
  const int row_stride = 3 * H * Dh;       // qkv row stride (elements)
  const int o_stride = H * Dh;
  const long qkv_base = (long)b * S * row_stride + (long)h * Dh;
  const long o_base = (long)b * S * o_stride + (long)h * Dh;
  const short* Qg = QKV + qkv_base;                 // row s: Qg[s*row_stride + d]
  const short* Kg = QKV + qkv_base + H * Dh;
  const short* Vg = QKV + qkv_base + 2 * H * Dh;

  const int tid = threadIdx.x;
  const int wid = tid / DMX_WAVE;
  const int lane = tid % DMX_WAVE;

  const int SKP = Dh + 8;   // K row stride (elements)
  const int SVP = S + 8;    // Vt + P row stride
  extern __shared__ __attribute__((aligned(16))) short smem[];
  short* k_lds = smem;                         // [S][SKP]
  short* vt_lds = k_lds + S * SKP;             // [Dh][SVP]
  short* p_lds = vt_lds + Dh * SVP;            // [AM_WAVES][16][SVP]

  // ---- stage K (row-major, vector-8) and V transposed ----
  {
    const int total = S * Dh;
    for (int i = tid * 8; i < total; i += AM_WAVES * DMX_WAVE * 8) {
      const int s = i / Dh, d = i % Dh;
      short8v kv = *(const short8v*)(Kg + (long)s * row_stride + d);
      *(short8v*)(k_lds + s * SKP + d) = kv;
      short8v vv = *(const short8v*)(Vg + (long)s * row_stride + d);
#pragma unroll
      for (int j = 0; j < 8; ++j) vt_lds[(d + j) * SVP + s] = vv[j];
    }
  }
  __syncthreads();

  const int q0 = wid * 16;
  if (q0 >= S) return;                  // no barriers after this point
  short* my_p = p_lds + wid * 16 * SVP;

  const int JF = S / 16;                // j fragments (score cols / 16)
  const int KQK = Dh / 32;              // k-steps for QK^T
  const int KPV = S / 32;               // k-steps for PV
  const int NF = Dh / 16;               // output d fragments

  // ---- QK^T into acc_p[JF] ----
  f32x4 acc_p[8];                       // JF <= 8 (S <= 128)
#pragma unroll
  for (int f = 0; f < 8; ++f) acc_p[f] = {0.f, 0.f, 0.f, 0.f};
  for (int ks = 0; ks < KQK; ++ks) {
    // A-frag: Q[q0 + (lane&15)][ks*32 + (lane>>4)*8 ..+7] from global
    bf16x8 a = *(const bf16x8*)(
        Qg + (long)(q0 + (lane & 15)) * row_stride + ks * 32 + (lane >> 4) * 8);
    for (int f = 0; f < JF; ++f) {
      bf16x8 bfr = *(const bf16x8*)(
          k_lds + (f * 16 + (lane & 15)) * SKP + ks * 32 + (lane >> 4) * 8);
      acc_p[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfr, acc_p[f], 0, 0, 0);
    }
  }

  // ---- softmax (full row; acc layout: row=(lane>>4)*4+r, col=lane&15) ----
  float inv_sum[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    float m = -1e30f;
    for (int f = 0; f < JF; ++f) m = fmaxf(m, acc_p[f][r] * scale);
#pragma unroll
    for (int mask = 1; mask < 16; mask <<= 1)
      m = fmaxf(m, __shfl_xor(m, mask, 64));
    float sum = 0.f;
    for (int f = 0; f < JF; ++f) {
      const float e = __expf(acc_p[f][r] * scale - m);
      acc_p[f][r] = e;
      sum += e;
    }
#pragma unroll
    for (int mask = 1; mask < 16; mask <<= 1)
      sum += __shfl_xor(sum, mask, 64);
    inv_sum[r] = 1.f / sum;
  }

  // ---- write unnormalized P to LDS (bf16) ----
  // element P[row][col]: row = (lane>>4)*4+r, col = f*16 + (lane&15)
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = (lane >> 4) * 4 + r;
    for (int f = 0; f < JF; ++f)
      my_p[row * SVP + f * 16 + (lane & 15)] = f32_to_bf16(acc_p[f][r]);
  }
  // wave-local LDS visibility (no cross-wave sharing of my_p)
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

  // ---- PV into acc_o[NF] ----
  f32x4 acc_o[4];                       // NF <= 4 (Dh <= 64)
#pragma unroll
  for (int f = 0; f < 4; ++f) acc_o[f] = {0.f, 0.f, 0.f, 0.f};
  for (int ks = 0; ks < KPV; ++ks) {
    bf16x8 a = *(const bf16x8*)(
        my_p + (lane & 15) * SVP + ks * 32 + (lane >> 4) * 8);
    for (int f = 0; f < NF; ++f) {
      bf16x8 bfr = *(const bf16x8*)(
          vt_lds + (f * 16 + (lane & 15)) * SVP + ks * 32 + (lane >> 4) * 8);
      acc_o[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfr, acc_o[f], 0, 0, 0);
    }
  }

  // ---- epilogue: normalize rows, store O[q][d] ----
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int q = q0 + (lane >> 4) * 4 + r;
    for (int f = 0; f < NF; ++f) {
      const int d = f * 16 + (lane & 15);
      O[o_base + (long)q * o_stride + d] =
          f32_to_bf16(acc_o[f][r] * inv_sum[r]);
    }
  }
}

extern "C" void dmx_launch_attention_mfma_bf16(
    const void* QKV, void* O, int B, int S, int H, int Dh, float scale,
    hipStream_t stream) {
  const int SKP = Dh + 8, SVP = S + 8;
  const size_t lds =
      ((size_t)S * SKP + (size_t)Dh * SVP + (size_t)AM_WAVES * 16 * SVP) *
      sizeof(short);
  hipLaunchKernelGGL(dmx_attention_mfma_bf16, dim3(B * H),
                     dim3(AM_WAVES * DMX_WAVE), lds, stream,
                     (const short*)QKV, (short*)O, B, S, H, Dh, scale);
}
