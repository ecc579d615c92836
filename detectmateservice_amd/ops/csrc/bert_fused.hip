// Fused whole-model BERT-tiny forward: ONE workgroup per log line.
//
// The flagship detector config (BASELINE.json config 5) is tiny enough
// that the idiomatic MI355X mapping is a per-line megakernel: all
// activations for one line (S=64 tokens, hidden 128) live in LDS for the
// whole forward pass; the model's weights (~600 KB) are read through the
// XCD L2s and shared by every workgroup; HBM traffic is line bytes in +
// one f32 score out.
//
// v3 design notes (PMC-driven, profiles/*):
//  * v1 (layered kernels) was HBM-bound on inter-op traffic (74% GEMM).
//  * v2 (fused, 118 KiB arena, 1 block/CU) showed MemUnitStalled~0 and
//    VALUBusy 33%: bulk-synchronous phases at 1 block/CU leave the CU
//    idle at every barrier. This version shrinks the arena to ~70 KiB so
//    TWO blocks co-reside per CU - one block's barrier idle overlaps the
//    other block's compute. 8 waves/block (512 thr), VGPR<=128 keeps both
//    resident (4 waves/SIMD from two independent barrier domains).
//  * arena cuts: qkv buffer holds Q|K only (V written transposed to vt
//    during the qkv GEMM epilogue); FFN runs in two K=256 halves that
//    accumulate into x; the attention P tile aliases the (dead) Q|K area.
//  * pair-chunked weight fragments: one L2 B-fragment load feeds two
//    independent MFMA chains.
//
// Fixed geometry (asserted host-side): S=64 tokens, H=128, 2 heads
// (Dh=64), FFN=512, arbitrary layer count.
#include "common.h"

// Explicit LDS address-space typing: the noinline block_gemm receives the
// arena pointers as arguments, and a generic (flat) short* there makes
// hipcc emit flat_load for what are really LDS reads (35 flat_load in the
// .s before this change). AS(3) pointers guarantee ds_* lowering.
typedef __attribute__((address_space(3))) short lds_short;
typedef __attribute__((address_space(1))) const short glob_cshort;
typedef __attribute__((address_space(1))) const float glob_cfloat;
typedef __attribute__((address_space(3))) const short lds_cshort;
typedef __attribute__((address_space(3))) float lds_float;

#define BF_WAVES 8
#define BF_THREADS (BF_WAVES * DMX_WAVE)
#define BF_S 64
#define BF_H 128
#define BF_DH 64
#define BF_FFN 512

// LDS strides (elements); +8 pads keep rows 16-B aligned and bank-spread
#define XS (BF_H + 8)       // 136, x rows
// QKS at +16 (not +8): stride 136 dw = 8 mod 64 makes the FFN W2 /
// attention A-fragment ds_read_b128 groups conflict-free (PMC: FFN's
// 6.98B conflict cycles are 100% A-reads — probe ABL=2 zeroes them).
// Q|K (64x272=17408) and FFN halves still fit inside BUF_ELEMS (17920,
// sized by the P+O region), so this costs ZERO extra LDS. Measured
// NEUTRAL on wall (7.30M both ways): in this latency-bound regime the
// conflict cycles hide under the operand waits — kept because it is
// free and removes them from every future profile.
#define QKS (2 * BF_H + 16)  // 272, Q|K rows and FFN-half rows
#define VTS (BF_S + 8)      // 72, transposed-V and P rows
#define O_OFF (BF_WAVES * 16 * VTS)    // 9216: attn-out after the P tiles
#define BUF_ELEMS (O_OFF + BF_S * XS)  // 17920 = P+O (>= QK 17408)

// bf16 weight-blob element offsets (host packs identically: bert_tiny.py)
#define WB_TOK 0
#define WB_POS (WB_TOK + 259 * BF_H)
#define WB_LAYER0 (WB_POS + BF_S * BF_H)
#define LW_QKV 0
#define LW_WO (LW_QKV + 3 * BF_H * BF_H)
#define LW_W1 (LW_WO + BF_H * BF_H)
#define LW_W2 (LW_W1 + BF_FFN * BF_H)
#define LW_LN1G (LW_W2 + BF_H * BF_FFN)
#define LW_LN1B (LW_LN1G + BF_H)
#define LW_LN2G (LW_LN1B + BF_H)
#define LW_LN2B (LW_LN2G + BF_H)
#define LW_SIZE (LW_LN2B + BF_H)
// f32 blob: per layer [bqkv 384 | bo 128 | b1 512 | b2 128], then b_score
#define FB_BQKV 0
#define FB_BO (FB_BQKV + 3 * BF_H)
#define FB_B1 (FB_BO + BF_H)
#define FB_B2 (FB_B1 + BF_FFN)
#define FB_SIZE (FB_B2 + BF_H)

// ---- in-block GEMM: out = act(in_lds[64][K] @ Wt[N][WTS] + bias) ---------
// MODE 0: write out_lds[m][n]; MODE 1 (qkv): n<2H -> out (Q|K), else vt
// transposed; MODE 2: x[m][n] += v (residual-accumulate). ACT 1 = GELU.
// ABL (perf-ablation diagnostics, numerically wrong on purpose):
// bit 1 = B operand from an opaque register instead of the L2 weight
// load (removes vmcnt parks); bit 2 = A operand from a register instead
// of the LDS read (removes lgkmcnt parks). Epilogue stores stay, so the
// MFMA chains cannot be dead-code-eliminated (guide §5.4 rule 17).
// All pointers __restrict__: without it the epilogue's LDS stores
// between quads alias the activation reads for the analyzer (the
// pointers arrive as opaque noinline args), so the compiler re-reads
// every A fragment per quad (32 ds_reads for 32 MFMAs in the .s) and
// cannot hoist reads across the quad boundary. The caller guarantees
// in_lds never overlaps this GEMM's outputs.
template <int K, int N, int MODE, int ACT, int WTS, int ABL = 0>
static __device__ __attribute__((noinline)) void block_gemm(
    const lds_short* __restrict__ in_lds, int in_stride,
    const glob_cshort* __restrict__ Wt, const glob_cfloat* __restrict__ bias,
    lds_short* __restrict__ out_lds, int out_stride,
    lds_short* __restrict__ x_lds, lds_short* __restrict__ vt_lds, int wid,
    int lane) {
  bf16x8 zb;
  if constexpr (ABL != 0) {
#pragma unroll
    for (int j = 0; j < 8; ++j) zb[j] = (short)0x3f01;  // ~1.008 bf16
    asm volatile("" : "+v"(zb));  // opaque: keeps realistic data/clocks
  }
  constexpr int N16 = N / 16;
  constexpr int TOTAL = 4 * N16;
  constexpr int KS = K / 32;
  constexpr int FPW = TOTAL / BF_WAVES;
  static_assert(FPW >= 4 && FPW % 4 == 0, "quad-chunked assignment");
  // quad-chunk: each iteration owns a FULL fn column (all 4 m-fragments):
  // ONE L2 weight-fragment load feeds FOUR independent MFMA chains.
  // unroll 2 quads: the second quad's (independent) weight loads issue
  // under the first quad's MFMA chains, hiding the L2 latency that
  // dominated the per-quad cost (phase probe: GEMMs ~3.9 of 4.5 ms while
  // pure MFMA issue accounts for <10% of that)
#pragma unroll 2
  for (int ff = wid * FPW; ff < wid * FPW + FPW; ff += 4) {
    const int fn = ff >> 2;
    // bias up front: a bias load in the epilogue made the compiler wait
    // vmcnt(0) mid-GEMM, draining the weight-load pipeline with it
    const int n = fn * 16 + (lane & 15);
    const float bval = bias ? bias[n] : 0.f;
    f32x4 acc[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll 4
    for (int ks = 0; ks < KS; ++ks) {
      bf16x8 b;
      if constexpr (ABL & 1) {
        b = zb;
      } else {
        b = *(const __attribute__((address_space(1))) bf16x8*)(
            Wt + (long)(fn * 16 + (lane & 15)) * WTS + ks * 32 +
            (lane >> 4) * 8);
      }
#pragma unroll
      for (int fm = 0; fm < 4; ++fm) {
        bf16x8 a;
        if constexpr (ABL & 2) {
          a = zb;
        } else {
          a = *(const __attribute__((address_space(3))) bf16x8*)(
              in_lds + (fm * 16 + (lane & 15)) * in_stride + ks * 32 +
              (lane >> 4) * 8);
        }
        acc[fm] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[fm], 0, 0, 0);
      }
    }
#pragma unroll
    for (int fm = 0; fm < 4; ++fm) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = fm * 16 + (lane >> 4) * 4 + r;
        float v = acc[fm][r] + bval;
        if (MODE == 0) {
          if (ACT == 1) v = gelu_f32(v);
          out_lds[m * out_stride + n] = f32_to_bf16(v);
        } else if (MODE == 1) {
          if (n < 2 * BF_H) {
            out_lds[m * out_stride + n] = f32_to_bf16(v);
          } else {
            vt_lds[(n - 2 * BF_H) * VTS + m] = f32_to_bf16(v);  // V transposed
          }
        } else {  // MODE 2
          const float xv = bf16_to_f32(x_lds[m * XS + n]);
          x_lds[m * XS + n] = f32_to_bf16(v + xv);
        }
      }
    }
  }
}

// ---- in-block LayerNorm on x (post-LN) -----------------------------------
// All 8 of the wave's rows in ONE parallel pass: 8 lanes per row, 16
// elements per lane, 3-step shfl_xor reduction within each 8-lane row
// group. (The serial per-row loop with full-wave reductions measured
// +1.12 ms of the 5.2 ms step — as costly as the whole QKV GEMM.)
static __device__ __forceinline__ void block_layernorm(
    lds_short* x_lds, const short* __restrict__ gamma,
    const short* __restrict__ beta, int wid, int lane, float eps) {
  const int row = wid * 8 + (lane >> 3);
  const int c0 = (lane & 7) * 16;
  short8v va = *(const __attribute__((address_space(3))) short8v*)(x_lds + row * XS + c0);
  short8v vb = *(const __attribute__((address_space(3))) short8v*)(x_lds + row * XS + c0 + 8);
  float v[16];
  float sum = 0.f;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    v[j] = bf16_to_f32(va[j]);
    v[8 + j] = bf16_to_f32(vb[j]);
    sum += v[j] + v[8 + j];
  }
#pragma unroll
  for (int mask = 1; mask < 8; mask <<= 1) sum += __shfl_xor(sum, mask, 64);
  const float mean = sum / BF_H;
  float var = 0.f;
#pragma unroll
  for (int j = 0; j < 16; ++j) {
    const float d = v[j] - mean;
    var += d * d;
  }
#pragma unroll
  for (int mask = 1; mask < 8; mask <<= 1) var += __shfl_xor(var, mask, 64);
  const float rstd = rsqrtf(var / BF_H + eps);
  short8v ga = *(const short8v*)(gamma + c0);
  short8v gb = *(const short8v*)(gamma + c0 + 8);
  short8v ba = *(const short8v*)(beta + c0);
  short8v bb = *(const short8v*)(beta + c0 + 8);
  short8v oa, ob;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    oa[j] = f32_to_bf16((v[j] - mean) * rstd * bf16_to_f32(ga[j]) +
                        bf16_to_f32(ba[j]));
    ob[j] = f32_to_bf16((v[8 + j] - mean) * rstd * bf16_to_f32(gb[j]) +
                        bf16_to_f32(bb[j]));
  }
  *(__attribute__((address_space(3))) short8v*)(x_lds + row * XS + c0) = oa;
  *(__attribute__((address_space(3))) short8v*)(x_lds + row * XS + c0 + 8) = ob;
}

// Phase bits for the profiling probe (production uses PH_ALL; skipped
// phases cannot be dead-code-eliminated backwards because every phase
// ends in LDS stores).
#define PH_QKV 1
#define PH_ATTN 2
#define PH_PROJ 4
#define PH_LN 8
#define PH_FFN 16
#define PH_ALL 31

// Per-wave phase timing (diagnostic builds): stamps[k] = s_memrealtime at
// phase boundaries; slot pairs (work-done, barrier-crossed) separate each
// wave's compute time from its barrier wait. 100 MHz constant clock.
#define BF_NSTAMP 22

template <int PHASES, bool TIMED = false, int ABL = 0>
static __device__ __forceinline__ void bert_fused_body(
    const unsigned char* __restrict__ lines,  // [B, max_len]
    const int* __restrict__ start,            // [B] content span start
    const int* __restrict__ end,              // [B] content span end
    const short* __restrict__ wb,             // bf16 weight blob
    const float* __restrict__ fb,             // f32 bias blob
    float* __restrict__ scores,               // [B]
    int B, int max_len, int n_layers, float eps,
    unsigned long long* __restrict__ stamps_out = nullptr) {
  unsigned long long tstamp[TIMED ? BF_NSTAMP : 1];
  int tidx = 0;
#define BF_STAMP()                                                          \
  if constexpr (TIMED) {                                                    \
    if (tidx < BF_NSTAMP) tstamp[tidx++] = __builtin_amdgcn_s_memrealtime(); \
  }
  const int line = blockIdx.x;
  if (line >= B) return;
  const int tid = threadIdx.x;
  const int wid = tid / DMX_WAVE;
  const int lane = tid % DMX_WAVE;

  extern __shared__ __attribute__((aligned(16))) short smem_raw[];
  lds_short* smem = (lds_short*)smem_raw;
  lds_short* x_lds = smem;             // [64][XS]
  lds_short* buf = x_lds + BF_S * XS;  // BUF_ELEMS: QK | P+O | FFN-half
  lds_short* vt = buf + BUF_ELEMS;     // [128][VTS]
  lds_float* red = (lds_float*)(vt + 2 * BF_DH * VTS);  // [128] pooling

  BF_STAMP();  // 0: kernel start
  // ---- embed: x[s][c] = tok_emb[byte+3 or 0][c] + pos_emb[s][c] ----
  {
    const int s0 = start[line], e0 = end[line];
    for (int i = tid * 8; i < BF_S * BF_H; i += BF_THREADS * 8) {
      const int s = i / BF_H, c = i % BF_H;
      int tok = 0;
      const int idx = s0 + s;
      if (idx < e0 && idx < max_len)
        tok = (int)lines[(long)line * max_len + idx] + 3;
      short8v te = *(const short8v*)(wb + WB_TOK + (long)tok * BF_H + c);
      short8v pe = *(const short8v*)(wb + WB_POS + (long)s * BF_H + c);
      short8v xv;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        xv[j] = f32_to_bf16(bf16_to_f32(te[j]) + bf16_to_f32(pe[j]));
      *(__attribute__((address_space(3))) short8v*)(x_lds + s * XS + c) = xv;
    }
  }
  BF_STAMP();  // 1: embed work done
  __syncthreads();
  BF_STAMP();  // 2: embed barrier crossed

  for (int layer = 0; layer < n_layers; ++layer) {
    const short* lw = wb + WB_LAYER0 + (long)layer * LW_SIZE;
    const float* lf = fb + (long)layer * FB_SIZE;

    // ---- qkv: Q|K -> buf[64][QKS], V -> vt transposed ----
    if (PHASES & PH_QKV)
      block_gemm<BF_H, 3 * BF_H, 1, 0, BF_H, ABL>(x_lds, XS, (glob_cshort*)(lw + LW_QKV),
                                             (glob_cfloat*)(lf + FB_BQKV), buf, QKS, x_lds, vt,
                                             wid, lane);
    BF_STAMP();  // qkv work done
    __syncthreads();
    BF_STAMP();  // qkv barrier crossed

    // ---- attention: wave = (head hh, 16 q-rows) ----
    if (PHASES & PH_ATTN) {
      const int hh = wid >> 2;
      const int q0 = (wid & 3) * 16;
      const float scale = 0.125f;  // 1/sqrt(64)

      f32x4 acc_p[4];
#pragma unroll
      for (int f = 0; f < 4; ++f) acc_p[f] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ks = 0; ks < BF_DH / 32; ++ks) {
        bf16x8 a = *(const __attribute__((address_space(3))) bf16x8*)(
            buf + (q0 + (lane & 15)) * QKS + hh * BF_DH + ks * 32 +
            (lane >> 4) * 8);
#pragma unroll
        for (int f = 0; f < 4; ++f) {
          bf16x8 b = *(const __attribute__((address_space(3))) bf16x8*)(
              buf + (f * 16 + (lane & 15)) * QKS + BF_H + hh * BF_DH +
              ks * 32 + (lane >> 4) * 8);
          acc_p[f] =
              __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc_p[f], 0, 0, 0);
        }
      }
      float inv_sum[4];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float m = -1e30f;
#pragma unroll
        for (int f = 0; f < 4; ++f) m = fmaxf(m, acc_p[f][r] * scale);
#pragma unroll
        for (int mask = 1; mask < 16; mask <<= 1)
          m = fmaxf(m, __shfl_xor(m, mask, 64));
        float sum = 0.f;
#pragma unroll
        for (int f = 0; f < 4; ++f) {
          const float e = __expf(acc_p[f][r] * scale - m);
          acc_p[f][r] = e;
          sum += e;
        }
#pragma unroll
        for (int mask = 1; mask < 16; mask <<= 1)
          sum += __shfl_xor(sum, mask, 64);
        inv_sum[r] = 1.f / sum;
      }
      // P tiles alias the Q|K area: every wave must be done reading Q/K
      __syncthreads();
      lds_short* my_p = buf + wid * 16 * VTS;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = (lane >> 4) * 4 + r;
#pragma unroll
        for (int f = 0; f < 4; ++f)
          my_p[row * VTS + f * 16 + (lane & 15)] = f32_to_bf16(acc_p[f][r]);
      }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");  // wave-local P

      f32x4 acc_o[4];
#pragma unroll
      for (int f = 0; f < 4; ++f) acc_o[f] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ks = 0; ks < BF_S / 32; ++ks) {
        bf16x8 a = *(const __attribute__((address_space(3))) bf16x8*)(
            my_p + (lane & 15) * VTS + ks * 32 + (lane >> 4) * 8);
#pragma unroll
        for (int f = 0; f < 4; ++f) {
          bf16x8 b = *(const __attribute__((address_space(3))) bf16x8*)(
              vt + (hh * BF_DH + f * 16 + (lane & 15)) * VTS + ks * 32 +
              (lane >> 4) * 8);
          acc_o[f] =
              __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc_o[f], 0, 0, 0);
        }
      }
      // attn out -> buf[O_OFF + q*XS + hh*64 + d] (disjoint from P tiles)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int q = q0 + (lane >> 4) * 4 + r;
        const float inv = __shfl(inv_sum[r], (lane >> 4) * 4 + r, 64);
#pragma unroll
        for (int f = 0; f < 4; ++f) {
          const int d = f * 16 + (lane & 15);
          buf[O_OFF + q * XS + hh * BF_DH + d] =
              f32_to_bf16(acc_o[f][r] * inv);
        }
      }
    }
    BF_STAMP();  // attention work done
    __syncthreads();
    BF_STAMP();  // attention barrier crossed

    // ---- proj: x += Wo(attn) ; LN1 ----
    if (PHASES & PH_PROJ)
      block_gemm<BF_H, BF_H, 2, 0, BF_H, ABL>(buf + O_OFF, XS, (glob_cshort*)(lw + LW_WO),
                                         (glob_cfloat*)(lf + FB_BO), nullptr, 0, x_lds,
                                         nullptr, wid, lane);
    __syncthreads();
    if (PHASES & PH_LN)
      block_layernorm(x_lds, lw + LW_LN1G, lw + LW_LN1B, wid, lane, eps);
    BF_STAMP();  // proj+LN1 work done
    __syncthreads();
    BF_STAMP();  // proj+LN1 barrier crossed

    // ---- FFN in two K=256 halves: buf = gelu(x@W1_h); x += buf@W2_h ----
    if (PHASES & PH_FFN)
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      block_gemm<BF_H, BF_FFN / 2, 0, 1, BF_H, ABL>(
          x_lds, XS, (glob_cshort*)(lw + LW_W1 + (long)h * (BF_FFN / 2) * BF_H),
          (glob_cfloat*)(lf + FB_B1 + h * (BF_FFN / 2)), buf, QKS, nullptr,
          nullptr, wid, lane);
      __syncthreads();
      // bias b2 added once (half 0); half 1 adds only the partial product
      block_gemm<BF_FFN / 2, BF_H, 2, 0, BF_FFN, ABL>(
          buf, QKS, (glob_cshort*)(lw + LW_W2 + (long)h * (BF_FFN / 2)),
          h == 0 ? (glob_cfloat*)(lf + FB_B2) : nullptr, nullptr, 0, x_lds,
          nullptr, wid, lane);
      __syncthreads();
    }
    if (PHASES & PH_LN)
      block_layernorm(x_lds, lw + LW_LN2G, lw + LW_LN2B, wid, lane, eps);
    BF_STAMP();  // ffn+LN2 work done
    __syncthreads();
    BF_STAMP();  // ffn+LN2 barrier crossed
  }

  // ---- pool (mean over S) + score head ----
  {
    if (tid < BF_H) {
      float s = 0.f;
      for (int row = 0; row < BF_S; ++row)
        s += bf16_to_f32(x_lds[row * XS + tid]);
      const float w =
          bf16_to_f32(wb[WB_LAYER0 + (long)n_layers * LW_SIZE + tid]);
      red[tid] = (s / BF_S) * w;
    }
    __syncthreads();
    if (wid == 0) {
      float v = red[lane] + red[lane + 64];
      v = warp_reduce_sum_f32(v);
      if (lane == 0)
        scores[line] = v + fb[(long)n_layers * FB_SIZE];  // b_score
    }
  }
  BF_STAMP();  // final
  if constexpr (TIMED) {
    if (lane == 0 && stamps_out != nullptr) {
      unsigned long long* dst =
          stamps_out + ((long)line * BF_WAVES + wid) * BF_NSTAMP;
      for (int k = 0; k < BF_NSTAMP; ++k)
        dst[k] = k < tidx ? tstamp[k] : 0ull;
    }
  }
#undef BF_STAMP
}

extern "C" __global__ __launch_bounds__(BF_THREADS, 4)
void dmx_bert_fused_bf16(const unsigned char* __restrict__ lines,
                         const int* __restrict__ start,
                         const int* __restrict__ end,
                         const short* __restrict__ wb,
                         const float* __restrict__ fb,
                         float* __restrict__ scores, int B, int max_len,
                         int n_layers, float eps) {
  bert_fused_body<PH_ALL>(lines, start, end, wb, fb, scores, B, max_len,
                          n_layers, eps);
}

// probe variants (in-kernel phase ablation; guide §5.4 rule 19: co-compiled
// variants can perturb codegen by a few % — read the deltas, not absolutes)
template <int PHASES, int ABL = 0>
__global__ __launch_bounds__(BF_THREADS, 4) void dmx_bert_fused_probe(
    const unsigned char* lines, const int* start, const int* end,
    const short* wb, const float* fb, float* scores, int B, int max_len,
    int n_layers, float eps) {
  bert_fused_body<PHASES, false, ABL>(lines, start, end, wb, fb, scores, B,
                                      max_len, n_layers, eps);
}

extern "C" __global__ __launch_bounds__(BF_THREADS, 4)
void dmx_bert_fused_timed(const unsigned char* lines, const int* start,
                          const int* end, const short* wb, const float* fb,
                          float* scores, int B, int max_len, int n_layers,
                          float eps, unsigned long long* stamps) {
  bert_fused_body<PH_ALL, true>(lines, start, end, wb, fb, scores, B,
                                max_len, n_layers, eps, stamps);
}

extern "C" void dmx_launch_bert_fused_timed(
    const void* lines, const void* start, const void* end, const void* wb,
    const void* fb, void* scores, void* stamps, int B, int max_len,
    int n_layers, float eps, hipStream_t stream) {
  const size_t lds =
      ((size_t)BF_S * XS + BUF_ELEMS + (size_t)2 * BF_DH * VTS) *
          sizeof(short) +
      128 * sizeof(float);
  hipFuncSetAttribute((const void*)dmx_bert_fused_timed,
                      hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);
  hipLaunchKernelGGL(dmx_bert_fused_timed, dim3(B), dim3(BF_THREADS), lds,
                     stream, (const unsigned char*)lines, (const int*)start,
                     (const int*)end, (const short*)wb, (const float*)fb,
                     (float*)scores, B, max_len, n_layers, eps,
                     (unsigned long long*)stamps);
}

extern "C" void dmx_launch_bert_fused_probe(
    const void* lines, const void* start, const void* end, const void* wb,
    const void* fb, void* scores, int B, int max_len, int n_layers,
    float eps, int phase_mask, hipStream_t stream) {
  const size_t lds =
      ((size_t)BF_S * XS + BUF_ELEMS + (size_t)2 * BF_DH * VTS) *
          sizeof(short) +
      128 * sizeof(float);
#define LAUNCH_PROBE(PH, AB)                                                 \
  case ((PH) | ((AB) << 8)):                                                 \
    hipFuncSetAttribute((const void*)dmx_bert_fused_probe<PH, AB>,           \
                        hipFuncAttributeMaxDynamicSharedMemorySize,          \
                        (int)lds);                                           \
    hipLaunchKernelGGL((dmx_bert_fused_probe<PH, AB>), dim3(B),              \
                       dim3(BF_THREADS), lds, stream,                        \
                       (const unsigned char*)lines, (const int*)start,       \
                       (const int*)end, (const short*)wb, (const float*)fb,  \
                       (float*)scores, B, max_len, n_layers, eps);           \
    break;
  switch (phase_mask) {
    LAUNCH_PROBE(0, 0)
    LAUNCH_PROBE(PH_QKV, 0)
    LAUNCH_PROBE(PH_QKV | PH_ATTN, 0)
    LAUNCH_PROBE(PH_QKV | PH_ATTN | PH_PROJ, 0)
    LAUNCH_PROBE(PH_QKV | PH_ATTN | PH_PROJ | PH_LN, 0)
    LAUNCH_PROBE(PH_ALL, 0)
    LAUNCH_PROBE(PH_FFN, 0)
    LAUNCH_PROBE(PH_LN, 0)
    // perf-ablation variants (numerically wrong by design):
    LAUNCH_PROBE(PH_ALL, 1)   // B operand constant (no L2 weight loads)
    LAUNCH_PROBE(PH_ALL, 2)   // A operand constant (no LDS activation reads)
    LAUNCH_PROBE(PH_ALL, 3)   // both constant (pure MFMA + epilogue)
    LAUNCH_PROBE(PH_FFN, 1)
    LAUNCH_PROBE(PH_FFN, 2)
    LAUNCH_PROBE(PH_FFN, 3)
    default:
      break;
  }
#undef LAUNCH_PROBE
}

extern "C" void dmx_launch_bert_fused_bf16(
    const void* lines, const void* start, const void* end, const void* wb,
    const void* fb, void* scores, int B, int max_len, int n_layers, float eps,
    hipStream_t stream) {
  const size_t lds =
      ((size_t)BF_S * XS + BUF_ELEMS + (size_t)2 * BF_DH * VTS) *
          sizeof(short) +
      128 * sizeof(float);
  static bool attr_set = false;
  if (!attr_set) {
    hipFuncSetAttribute((const void*)dmx_bert_fused_bf16,
                        hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);
    attr_set = true;
  }
  hipLaunchKernelGGL(dmx_bert_fused_bf16, dim3(B), dim3(BF_THREADS), lds,
                     stream, (const unsigned char*)lines, (const int*)start,
                     (const int*)end, (const short*)wb, (const float*)fb,
                     (float*)scores, B, max_len, n_layers, eps);
}
