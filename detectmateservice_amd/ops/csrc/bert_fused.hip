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

#define BF_WAVES 4
#define BF_THREADS (BF_WAVES * DMX_WAVE)
#define BF_S 64
#define BF_H 128
#define BF_DH 64
#define BF_FFN 512

// LDS strides (elements); +8 pads keep rows 16-B aligned and bank-spread
#define XS (BF_H + 8)       // 136, x rows
#define QKS (2 * BF_H + 8)  // 264, Q|K rows and FFN-half rows
#define VTS (BF_S + 8)      // 72, transposed-V and P rows
#define O_OFF (BF_WAVES * 32 * VTS)    // 9216: attn-out after the P tiles
#define BUF_ELEMS (BF_S * QKS + 1024)  // 17920 = max(QK, P+O, FFN-half)

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

// ---- cross-barrier weight prefetch (v5) ----------------------------------
// Round-1 PMC evidence (profiles/r08): SQ_WAIT_ANY ~ 18x SQ_BUSY — waves
// spend ~95% of residency parked, and five issue-side ablations were
// neutral. The wall is structural: every GEMM phase opened with a
// DEPENDENT chain of L2 weight-fragment loads issued AFTER the barrier
// that every wave crosses together, so the whole CU idles ~KS x 200
// cycles per phase before the first MFMA can retire. Weights are
// read-only, so those loads have NO hazard against the barrier — but the
// compiler cannot hoist loads across __syncthreads(). v5 does it by
// hand: each GEMM's full B-fragment set is issued into registers during
// the PREVIOUS phase and consumed after the barrier, so phases open
// MFMA-ready (the s_waitcnt lands ~a whole phase after issue). Costs
// VGPRs (fragments held across a phase): launch bounds drop to
// 2 waves/SIMD (<=256 VGPR, 1 block/CU) — round-1 measurements showed
// throughput FLAT in occupancy 2-6 waves/SIMD, so the co-residency this
// gives up was not buying anything the prefetch doesn't replace.
//
// Per wave a GEMM owns N/128 "quads" (quad = one fn column x 4
// m-fragments); fragments per wave = (N/128) * (K/32).
// PKS = fragments per quad prefetched across the barrier (the rest
// stream in-phase under MFMA cover): full prefetch of every GEMM spilled
// 17 VGPRs in-loop; phase-START latency only needs the first fragments
// resident, later ks-steps hide behind ~136 cycles of MFMA per step.
template <int K, int N, int PKS, int WTS>
static __device__ __forceinline__ void load_wfrags(
    const glob_cshort* __restrict__ Wt, int wid, int lane, bf16x8* out) {
  constexpr int NQ = N / (16 * BF_WAVES);
  // opaque lane: the per-lane address offsets are layer-invariant and
  // LICM otherwise hoists ~25 of them out of the layer loop, spilling
  // them across every phase (the guide's "recompute per block" pitfall);
  // recomputing costs a couple of VALU per phase.
  int ln = lane;
  asm volatile("" : "+v"(ln));
#pragma unroll
  for (int q = 0; q < NQ; ++q)
#pragma unroll
    for (int ks = 0; ks < PKS; ++ks)
      out[q * PKS + ks] = *(const __attribute__((address_space(1))) bf16x8*)(
          Wt + (long)((wid * NQ + q) * 16 + (ln & 15)) * WTS + ks * 32 +
          (ln >> 4) * 8);
}

// Biases ride along with the fragments: an in-phase bias load would make
// the compiler emit a vmcnt(0) drain in the GEMM epilogue, flushing the
// NEXT phase's just-issued prefetch loads with it.
template <int N>
static __device__ __forceinline__ void load_bias(
    const glob_cfloat* __restrict__ bias, int wid, int lane, float* out) {
  constexpr int NQ = N / (16 * BF_WAVES);
  int ln = lane;
  asm volatile("" : "+v"(ln));
#pragma unroll
  for (int q = 0; q < NQ; ++q)
    out[q] = bias ? bias[(wid * NQ + q) * 16 + (ln & 15)] : 0.f;
}

// ---- in-block GEMM: out = act(in_lds[64][K] @ W + bias) ------------------
// Weight fragments arrive PRELOADED in registers (load_wfrags, issued
// before the preceding barrier). MODE 0: write out_lds[m][n]; MODE 1
// (qkv): n<2H -> out (Q|K), else vt transposed; MODE 2: x[m][n] += v
// (residual-accumulate). ACT 1 = GELU.
template <int K, int N, int PKS, int MODE, int ACT, int WTS>
static __device__ __forceinline__ void block_gemm_pre(
    const lds_short* in_lds, int in_stride, const bf16x8* w,
    const glob_cshort* __restrict__ Wt, const float* bias_pre,
    lds_short* out_lds, int out_stride, lds_short* x_lds, lds_short* vt_lds,
    int wid, int lane) {
  constexpr int KS = K / 32;
  constexpr int NQ = N / (16 * BF_WAVES);
#pragma unroll
  for (int q = 0; q < NQ; ++q) {
    // cap register pressure: without this fence the fully-unrolled quad
    // loop interleaves every quad's 16 accumulators (192 VGPR spills at
    // NQ=6). Mask 0x120 (VMEM_READ|DS_READ) still lets the NEXT quad's
    // weight/activation loads hoist under THIS quad's MFMAs.
    if (q > 0) __builtin_amdgcn_sched_barrier(0x120);
    const int fn = wid * NQ + q;
    f32x4 acc[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};
    // prefetched head fragments
#pragma unroll
    for (int ks = 0; ks < PKS; ++ks) {
      const bf16x8 b = w[q * PKS + ks];
#pragma unroll
      for (int fm = 0; fm < 4; ++fm) {
        bf16x8 a = *(const __attribute__((address_space(3))) bf16x8*)(
            in_lds + (fm * 16 + (lane & 15)) * in_stride + ks * 32 +
            (lane >> 4) * 8);
        acc[fm] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[fm], 0, 0, 0);
      }
    }
    // in-phase tail: loads issue under the head's MFMA chains
    int ln = lane;
    asm volatile("" : "+v"(ln));
#pragma unroll
    for (int ks = PKS; ks < KS; ++ks) {
      const bf16x8 b = *(const __attribute__((address_space(1))) bf16x8*)(
          Wt + (long)(fn * 16 + (ln & 15)) * WTS + ks * 32 +
          (ln >> 4) * 8);
#pragma unroll
      for (int fm = 0; fm < 4; ++fm) {
        bf16x8 a = *(const __attribute__((address_space(3))) bf16x8*)(
            in_lds + (fm * 16 + (lane & 15)) * in_stride + ks * 32 +
            (lane >> 4) * 8);
        acc[fm] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[fm], 0, 0, 0);
      }
    }
    const int n = fn * 16 + (lane & 15);
    const float bval = bias_pre[q];
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
  // 4 waves x 2 passes x 8 rows = the 64 rows (8 lanes per row, 16
  // elements per lane, 3-step shfl_xor reduction per 8-lane row group)
#pragma unroll
  for (int rr = 0; rr < BF_S / (BF_WAVES * 8); ++rr) {
  const int row = wid * (BF_S / BF_WAVES) + rr * 8 + (lane >> 3);
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

template <int PHASES, bool TIMED = false>
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
  // layer-0 qkv weight fragments: issued HERE so they are in flight
  // across the embed barrier (load_wfrags comment above)
  bf16x8 wq[6];
  float bq[6];
  load_wfrags<BF_H, 3 * BF_H, 1, BF_H>(
      (glob_cshort*)(wb + WB_LAYER0 + LW_QKV), wid, lane, wq);
  load_bias<3 * BF_H>((glob_cfloat*)(fb + FB_BQKV), wid, lane, bq);
  __syncthreads();
  BF_STAMP();  // 2: embed barrier crossed

  for (int layer = 0; layer < n_layers; ++layer) {
    const short* lw = wb + WB_LAYER0 + (long)layer * LW_SIZE;
    const float* lf = fb + (long)layer * FB_SIZE;

    // ---- qkv: Q|K -> buf[64][QKS], V -> vt transposed ----
    if (PHASES & PH_QKV)
      block_gemm_pre<BF_H, 3 * BF_H, 1, 1, 0, BF_H>(
          x_lds, XS, wq, (glob_cshort*)(lw + LW_QKV), bq, buf, QKS, x_lds,
          vt, wid, lane);
    BF_STAMP();  // qkv work done
    // proj weights ride across the qkv barrier + the whole attention
    // phase (attention reads no global weights)
    bf16x8 wpr[8];
    float bpr[2];
    load_wfrags<BF_H, BF_H, 4, BF_H>((glob_cshort*)(lw + LW_WO), wid, lane,
                                     wpr);
    load_bias<BF_H>((glob_cfloat*)(lf + FB_BO), wid, lane, bpr);
    __syncthreads();
    BF_STAMP();  // qkv barrier crossed

    // ---- attention: wave = (head hh, 32 q-rows as two 16-row groups) ----
    if (PHASES & PH_ATTN) {
      const int hh = wid >> 1;
      const int q0 = (wid & 1) * 32;
      const float scale = 0.125f;  // 1/sqrt(64)

      // both groups' QK^T + softmax BEFORE the P-alias barrier (every
      // wave must be done reading Q/K before any P store)
      f32x4 acc_p[2][4];
      float inv_sum[2][4];
#pragma unroll
      for (int g = 0; g < 2; ++g) {
#pragma unroll
        for (int f = 0; f < 4; ++f) acc_p[g][f] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int ks = 0; ks < BF_DH / 32; ++ks) {
          bf16x8 a = *(const __attribute__((address_space(3))) bf16x8*)(
              buf + (q0 + g * 16 + (lane & 15)) * QKS + hh * BF_DH + ks * 32 +
              (lane >> 4) * 8);
#pragma unroll
          for (int f = 0; f < 4; ++f) {
            bf16x8 b = *(const __attribute__((address_space(3))) bf16x8*)(
                buf + (f * 16 + (lane & 15)) * QKS + BF_H + hh * BF_DH +
                ks * 32 + (lane >> 4) * 8);
            acc_p[g][f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a, b, acc_p[g][f], 0, 0, 0);
          }
        }
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float m = -1e30f;
#pragma unroll
          for (int f = 0; f < 4; ++f) m = fmaxf(m, acc_p[g][f][r] * scale);
#pragma unroll
          for (int mask = 1; mask < 16; mask <<= 1)
            m = fmaxf(m, __shfl_xor(m, mask, 64));
          float sum = 0.f;
#pragma unroll
          for (int f = 0; f < 4; ++f) {
            const float e = __expf(acc_p[g][f][r] * scale - m);
            acc_p[g][f][r] = e;
            sum += e;
          }
#pragma unroll
          for (int mask = 1; mask < 16; mask <<= 1)
            sum += __shfl_xor(sum, mask, 64);
          inv_sum[g][r] = 1.f / sum;
        }
      }
      // P tiles alias the Q|K area: every wave must be done reading Q/K
      __syncthreads();
#pragma unroll
      for (int g = 0; g < 2; ++g) {
        lds_short* my_p = buf + (wid * 32 + g * 16) * VTS;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = (lane >> 4) * 4 + r;
#pragma unroll
          for (int f = 0; f < 4; ++f)
            my_p[row * VTS + f * 16 + (lane & 15)] =
                f32_to_bf16(acc_p[g][f][r]);
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
            acc_o[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a, b, acc_o[f], 0, 0, 0);
          }
        }
        // attn out -> buf[O_OFF + q*XS + hh*64 + d] (disjoint from P tiles)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int q = q0 + g * 16 + (lane >> 4) * 4 + r;
          const float inv = __shfl(inv_sum[g][r], (lane >> 4) * 4 + r, 64);
#pragma unroll
          for (int f = 0; f < 4; ++f) {
            const int d = f * 16 + (lane & 15);
            buf[O_OFF + q * XS + hh * BF_DH + d] =
                f32_to_bf16(acc_o[f][r] * inv);
          }
        }
      }
    }
    BF_STAMP();  // attention work done
    // FFN W1 half-0 fragments cross the attn barrier + proj + LN1
    bf16x8 w1[8];
    float b1[4];
    load_wfrags<BF_H, BF_FFN / 2, 2, BF_H>((glob_cshort*)(lw + LW_W1), wid,
                                           lane, w1);
    load_bias<BF_FFN / 2>((glob_cfloat*)(lf + FB_B1), wid, lane, b1);
    __syncthreads();
    BF_STAMP();  // attention barrier crossed

    // ---- proj: x += Wo(attn) ; LN1 ----
    if (PHASES & PH_PROJ)
      block_gemm_pre<BF_H, BF_H, 4, 2, 0, BF_H>(
          buf + O_OFF, XS, wpr, (glob_cshort*)(lw + LW_WO), bpr, nullptr, 0,
          x_lds, nullptr, wid, lane);
    // W2 half-0 fragments cross the proj barrier + LN1 + W1h0
    bf16x8 w2[8];
    float b2[2];
    load_wfrags<BF_FFN / 2, BF_H, 4, BF_FFN>((glob_cshort*)(lw + LW_W2), wid,
                                             lane, w2);
    load_bias<BF_H>((glob_cfloat*)(lf + FB_B2), wid, lane, b2);
    __syncthreads();
    if (PHASES & PH_LN)
      block_layernorm(x_lds, lw + LW_LN1G, lw + LW_LN1B, wid, lane, eps);
    BF_STAMP();  // proj+LN1 work done
    __syncthreads();
    BF_STAMP();  // proj+LN1 barrier crossed

    // ---- FFN in two K=256 halves: buf = gelu(x@W1_h); x += buf@W2_h ----
    if (PHASES & PH_FFN) {
      block_gemm_pre<BF_H, BF_FFN / 2, 2, 0, 1, BF_H>(
          x_lds, XS, w1, (glob_cshort*)(lw + LW_W1), b1, buf, QKS, nullptr,
          nullptr, wid, lane);
      // W1 half-1 fragments reuse w1's registers (h0 consumed above)
      load_wfrags<BF_H, BF_FFN / 2, 2, BF_H>(
          (glob_cshort*)(lw + LW_W1 + (long)(BF_FFN / 2) * BF_H), wid, lane,
          w1);
      load_bias<BF_FFN / 2>((glob_cfloat*)(lf + FB_B1 + BF_FFN / 2), wid,
                            lane, b1);
      __syncthreads();
      // bias b2 added once (half 0); half 1 adds only the partial product
      block_gemm_pre<BF_FFN / 2, BF_H, 4, 2, 0, BF_FFN>(
          buf, QKS, w2, (glob_cshort*)(lw + LW_W2), b2, nullptr, 0, x_lds,
          nullptr, wid, lane);
      load_wfrags<BF_FFN / 2, BF_H, 4, BF_FFN>(
          (glob_cshort*)(lw + LW_W2 + (BF_FFN / 2)), wid, lane, w2);
      b2[0] = b2[1] = 0.f;  // half 1 adds only the partial product
      __syncthreads();
      block_gemm_pre<BF_H, BF_FFN / 2, 2, 0, 1, BF_H>(
          x_lds, XS, w1,
          (glob_cshort*)(lw + LW_W1 + (long)(BF_FFN / 2) * BF_H), b1, buf,
          QKS, nullptr, nullptr, wid, lane);
      __syncthreads();
      block_gemm_pre<BF_FFN / 2, BF_H, 4, 2, 0, BF_FFN>(
          buf, QKS, w2, (glob_cshort*)(lw + LW_W2 + (BF_FFN / 2)), b2,
          nullptr, 0, x_lds, nullptr, wid, lane);
    }
    // next layer's qkv fragments cross the FFN-tail barrier + LN2
    if (layer + 1 < n_layers) {
      load_wfrags<BF_H, 3 * BF_H, 1, BF_H>(
          (glob_cshort*)(lw + LW_SIZE + LW_QKV), wid, lane, wq);
      load_bias<3 * BF_H>((glob_cfloat*)(lf + FB_SIZE + FB_BQKV), wid, lane,
                          bq);
    }
    __syncthreads();
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

extern "C" __global__ __launch_bounds__(BF_THREADS, 2)
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
template <int PHASES>
__global__ __launch_bounds__(BF_THREADS, 2) void dmx_bert_fused_probe(
    const unsigned char* lines, const int* start, const int* end,
    const short* wb, const float* fb, float* scores, int B, int max_len,
    int n_layers, float eps) {
  bert_fused_body<PHASES>(lines, start, end, wb, fb, scores, B, max_len,
                          n_layers, eps);
}

extern "C" __global__ __launch_bounds__(BF_THREADS, 2)
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
#define LAUNCH_PROBE(MASK)                                                   \
  case MASK:                                                                 \
    hipFuncSetAttribute((const void*)dmx_bert_fused_probe<MASK>,             \
                        hipFuncAttributeMaxDynamicSharedMemorySize,          \
                        (int)lds);                                           \
    hipLaunchKernelGGL(dmx_bert_fused_probe<MASK>, dim3(B),                  \
                       dim3(BF_THREADS), lds, stream,                        \
                       (const unsigned char*)lines, (const int*)start,       \
                       (const int*)end, (const short*)wb, (const float*)fb,  \
                       (float*)scores, B, max_len, n_layers, eps);           \
    break;
  switch (phase_mask) {
    LAUNCH_PROBE(0)
    LAUNCH_PROBE(PH_QKV)
    LAUNCH_PROBE(PH_QKV | PH_ATTN)
    LAUNCH_PROBE(PH_QKV | PH_ATTN | PH_PROJ)
    LAUNCH_PROBE(PH_QKV | PH_ATTN | PH_PROJ | PH_LN)
    LAUNCH_PROBE(PH_ALL)
    LAUNCH_PROBE(PH_FFN)
    LAUNCH_PROBE(PH_LN)
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
