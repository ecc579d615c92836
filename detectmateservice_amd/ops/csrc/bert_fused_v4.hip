// Fused BERT-tiny forward v4: OCCUPANCY variant — 3 blocks per CU.
//
// PMC evidence (profiles/r08): the v3 kernel's waves spend ~18x their
// busy cycles WAITING (SQ_WAIT_ANY / SQ_BUSY), and every issue-side
// ablation (GELU removal, shared weight loads, fewer epilogue ops)
// measured neutral — the kernel runs at the latency of its dependency
// chain and the lever is TLP: more co-resident blocks. v3's 70 KiB
// arena caps residency at 2 blocks/CU; this variant restructures the
// layer loop so the arena is ~44 KiB -> THREE blocks/CU (24 waves, 6
// waves/SIMD), at the cost of more phases/barriers (which the
// ablations showed are cheap):
//   * attention + QKV run PER HEAD (2 heads): only one head's Q|K|V^T
//     is ever resident;
//   * the P score tile aliases the (dead) K columns of the head buffer,
//     and the attention output O overwrites P in turn;
//   * the output projection accumulates per head (K-sliced Wo);
//   * the FFN runs in four K=128 quarters through the SAME head buffer.
// VGPR budget: 6 waves/SIMD needs <=85 VGPRs (-> __launch_bounds__
// (512, 3), no unroll-2 pipelining — TLP replaces ILP).
//
// Geometry fixed as v3: S=64, H=128, 2 heads (Dh=64), FFN=512.
#include "common.h"

typedef __attribute__((address_space(3))) short v4_lds_short;
typedef __attribute__((address_space(1))) const short v4_glob_cshort;
typedef __attribute__((address_space(1))) const float v4_glob_cfloat;
typedef __attribute__((address_space(3))) float v4_lds_float;

#define V4_WAVES 8
#define V4_THREADS (V4_WAVES * DMX_WAVE)
#define V4_S 64
#define V4_H 128
#define V4_DH 64
#define V4_FFN 512

// LDS strides/areas (shorts)
#define V4_XS 136                 // x rows: 128 + 8 pad
#define V4_BS 136                 // head buf rows: Q|K (64+64+8s... 136)
#define V4_VTS 72                 // V^T / P / O rows: 64 + 8 pad
#define V4_X_ELEMS (V4_S * V4_XS)       // 8704
#define V4_B_ELEMS (V4_S * V4_BS)       // 8704
#define V4_VT_ELEMS (V4_DH * V4_VTS)    // 4608
// f32 scratch: row max/sum halves for the split softmax (2*64 each)
#define V4_F32_ELEMS 256

// weight blob offsets (identical to bert_fused.hip / bert_tiny.py)
#define WB_TOK 0
#define WB_POS (WB_TOK + 259 * V4_H)
#define WB_LAYER0 (WB_POS + V4_S * V4_H)
#define LW_QKV 0
#define LW_WO (LW_QKV + 3 * V4_H * V4_H)
#define LW_W1 (LW_WO + V4_H * V4_H)
#define LW_W2 (LW_W1 + V4_FFN * V4_H)
#define LW_LN1G (LW_W2 + V4_H * V4_FFN)
#define LW_LN1B (LW_LN1G + V4_H)
#define LW_LN2G (LW_LN1B + V4_H)
#define LW_LN2B (LW_LN2G + V4_H)
#define LW_SIZE (LW_LN2B + V4_H)
#define FB_BQKV 0
#define FB_BO (FB_BQKV + 3 * V4_H)
#define FB_B1 (FB_BO + V4_H)
#define FB_B2 (FB_B1 + V4_FFN)
#define FB_SIZE (FB_B2 + V4_H)

// ---- generic in-block GEMM, N=128 per call (proj / FFN quarters) --------
// out = act(in[64][K] @ Wt[N=128 slice][WTS] + bias); one fn column per
// wave (quad-chunk: one weight fragment feeds 4 MFMA chains).
// MODE 0: write out_lds; MODE 2: x_lds residual-accumulate.
template <int K, int MODE, int ACT, int WTS>
static __device__ __attribute__((noinline)) void v4_gemm128(
    const v4_lds_short* in_lds, int in_stride,
    v4_glob_cshort* __restrict__ Wt, v4_glob_cfloat* __restrict__ bias,
    v4_lds_short* out_lds, int out_stride, v4_lds_short* x_lds, int wid,
    int lane) {
  constexpr int KS = K / 32;
  const int fn = wid;  // 8 waves == 8 column tiles of 16
  f32x4 acc[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
  for (int ks = 0; ks < KS; ++ks) {
    bf16x8 b = *(const __attribute__((address_space(1))) bf16x8*)(
        Wt + (long)(fn * 16 + (lane & 15)) * WTS + ks * 32 + (lane >> 4) * 8);
#pragma unroll
    for (int fm = 0; fm < 4; ++fm) {
      bf16x8 a = *(const __attribute__((address_space(3))) bf16x8*)(
          in_lds + (fm * 16 + (lane & 15)) * in_stride + ks * 32 +
          (lane >> 4) * 8);
      acc[fm] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[fm], 0, 0, 0);
    }
  }
  const int n = fn * 16 + (lane & 15);
  const float bval = bias ? bias[n] : 0.f;
#pragma unroll
  for (int fm = 0; fm < 4; ++fm) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = fm * 16 + (lane >> 4) * 4 + r;
      float v = acc[fm][r] + bval;
      if (MODE == 0) {
        if (ACT == 1) v = gelu_f32(v);
        out_lds[m * out_stride + n] = f32_to_bf16(v);
      } else {  // MODE 2
        const float xv = bf16_to_f32(x_lds[m * V4_XS + n]);
        x_lds[m * V4_XS + n] = f32_to_bf16(v + xv);
      }
    }
  }
}

// ---- per-head QKV GEMM ---------------------------------------------------
// N=192 columns (Qh|Kh|Vh), weight rows scattered in wqkv_t as
// [q(128) | k(128) | v(128)] with the head's 64-row slice of each.
// 12 column tiles -> 24 half-columns (2 m-frags each) over 8 waves.
static __device__ __attribute__((noinline)) void v4_qkv_head(
    const v4_lds_short* x_lds, v4_glob_cshort* __restrict__ Wqkv,
    v4_glob_cfloat* __restrict__ bqkv, v4_lds_short* hbuf,
    v4_lds_short* vt, int head, int wid, int lane) {
  for (int hc = wid * 3; hc < wid * 3 + 3; ++hc) {
    const int fn = hc >> 1;           // column tile 0..11
    const int fm0 = (hc & 1) * 2;     // 2 m-fragments per half-column
    const int seg = fn >> 2;          // 0=Q 1=K 2=V
    const int segcol = (fn & 3) * 16 + (lane & 15);  // 0..63 within segment
    const long wrow = (long)(seg * V4_H + head * V4_DH + segcol);
    f32x4 acc[2];
#pragma unroll
    for (int i = 0; i < 2; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int ks = 0; ks < V4_H / 32; ++ks) {
      bf16x8 b = *(const __attribute__((address_space(1))) bf16x8*)(
          Wqkv + wrow * V4_H + ks * 32 + (lane >> 4) * 8);
#pragma unroll
      for (int f = 0; f < 2; ++f) {
        bf16x8 a = *(const __attribute__((address_space(3))) bf16x8*)(
            x_lds + ((fm0 + f) * 16 + (lane & 15)) * V4_XS + ks * 32 +
            (lane >> 4) * 8);
        acc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[f], 0, 0, 0);
      }
    }
    const float bval = bqkv[seg * V4_H + head * V4_DH + segcol];
#pragma unroll
    for (int f = 0; f < 2; ++f) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = (fm0 + f) * 16 + (lane >> 4) * 4 + r;
        const float v = acc[f][r] + bval;
        if (seg < 2) {  // Q cols 0..63, K cols 64..127 of hbuf
          hbuf[m * V4_BS + seg * V4_DH + segcol] = f32_to_bf16(v);
        } else {        // V transposed: vt[d][m]
          vt[segcol * V4_VTS + m] = f32_to_bf16(v);
        }
      }
    }
  }
}

// ---- per-head attention --------------------------------------------------
// wave = (mt = wid&3: 16 q rows, nh = wid>>1&? ...) -> split: mt = wid & 3,
// nh = wid >> 2 (32-column half). Softmax rows span both halves -> two
// tiny LDS reductions (smax/ssum).
static __device__ __attribute__((noinline)) void v4_attention_head(
    v4_lds_short* hbuf, v4_lds_short* vt, v4_lds_float* smax,
    v4_lds_float* ssum, int wid, int lane) {
  const int mt = wid & 3;
  const int nh = wid >> 2;
  const int q0 = mt * 16;
  const float scale = 0.125f;  // 1/sqrt(64)
  // P = Q @ K^T : 16 x 32 tile
  f32x4 acc_p[2];
#pragma unroll
  for (int f = 0; f < 2; ++f) acc_p[f] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
  for (int ks = 0; ks < V4_DH / 32; ++ks) {
    bf16x8 a = *(const __attribute__((address_space(3))) bf16x8*)(
        hbuf + (q0 + (lane & 15)) * V4_BS + ks * 32 + (lane >> 4) * 8);
#pragma unroll
    for (int f = 0; f < 2; ++f) {
      bf16x8 b = *(const __attribute__((address_space(3))) bf16x8*)(
          hbuf + (nh * 32 + f * 16 + (lane & 15)) * V4_BS + V4_DH + ks * 32 +
          (lane >> 4) * 8);
      acc_p[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc_p[f], 0, 0, 0);
    }
  }
  // partial row max over this 32-col half -> smax[row*2 + nh]
  float pmax[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    float m = fmaxf(acc_p[0][r], acc_p[1][r]) * scale;
#pragma unroll
    for (int mask = 1; mask < 16; mask <<= 1)
      m = fmaxf(m, __shfl_xor(m, mask, 64));
    pmax[r] = m;
    if ((lane & 15) == 0) {
      const int row = q0 + (lane >> 4) * 4 + r;
      smax[row * 2 + nh] = m;
    }
  }
  __syncthreads();
  // combined max, exp, partial sums -> ssum[row*2 + nh]
  float inv_col_scale[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = q0 + (lane >> 4) * 4 + r;
    const float m = fmaxf(smax[row * 2], smax[row * 2 + 1]);
    float s = 0.f;
#pragma unroll
    for (int f = 0; f < 2; ++f) {
      const float e = __expf(acc_p[f][r] * scale - m);
      acc_p[f][r] = e;
      s += e;
    }
#pragma unroll
    for (int mask = 1; mask < 16; mask <<= 1) s += __shfl_xor(s, mask, 64);
    if ((lane & 15) == 0) ssum[row * 2 + nh] = s;
  }
  __syncthreads();
  // P tile (un-normalized exps) -> K-alias area: p[m][k] = hbuf[m*BS+64+k]
  // (K columns are dead: every wave finished its QK MFMAs before the
  // first barrier above)
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = q0 + (lane >> 4) * 4 + r;
    inv_col_scale[r] = 1.f / (ssum[row * 2] + ssum[row * 2 + 1]);
#pragma unroll
    for (int f = 0; f < 2; ++f)
      hbuf[row * V4_BS + V4_DH + nh * 32 + f * 16 + (lane & 15)] =
          f32_to_bf16(acc_p[f][r]);
  }
  __syncthreads();
  // O = P @ V^T : 16 x 32 tile (d-half nh); K = S = 64
  f32x4 acc_o[2];
#pragma unroll
  for (int f = 0; f < 2; ++f) acc_o[f] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
  for (int ks = 0; ks < V4_S / 32; ++ks) {
    bf16x8 a = *(const __attribute__((address_space(3))) bf16x8*)(
        hbuf + (q0 + (lane & 15)) * V4_BS + V4_DH + ks * 32 + (lane >> 4) * 8);
#pragma unroll
    for (int f = 0; f < 2; ++f) {
      bf16x8 b = *(const __attribute__((address_space(3))) bf16x8*)(
          vt + (nh * 32 + f * 16 + (lane & 15)) * V4_VTS + ks * 32 +
          (lane >> 4) * 8);
      acc_o[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc_o[f], 0, 0, 0);
    }
  }
  __syncthreads();
  // O (normalized) overwrites P in the alias area: o[m][d]=hbuf[m*BS+64+d]
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = q0 + (lane >> 4) * 4 + r;
#pragma unroll
    for (int f = 0; f < 2; ++f) {
      const int d = nh * 32 + f * 16 + (lane & 15);
      hbuf[row * V4_BS + V4_DH + d] =
          f32_to_bf16(acc_o[f][r] * inv_col_scale[r]);
    }
  }
}

// ---- LayerNorm (identical structure to v3) -------------------------------
static __device__ __attribute__((noinline)) void v4_layernorm(
    v4_lds_short* x_lds, const short* __restrict__ gamma,
    const short* __restrict__ beta, int wid, int lane, float eps) {
  const int row = wid * 8 + (lane >> 3);
  const int c0 = (lane & 7) * 16;
  short8v va = *(const __attribute__((address_space(3))) short8v*)(x_lds + row * V4_XS + c0);
  short8v vb = *(const __attribute__((address_space(3))) short8v*)(x_lds + row * V4_XS + c0 + 8);
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
  const float mean = sum / V4_H;
  float var = 0.f;
#pragma unroll
  for (int j = 0; j < 16; ++j) {
    const float d = v[j] - mean;
    var += d * d;
  }
#pragma unroll
  for (int mask = 1; mask < 8; mask <<= 1) var += __shfl_xor(var, mask, 64);
  const float rstd = rsqrtf(var / V4_H + eps);
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
  *(__attribute__((address_space(3))) short8v*)(x_lds + row * V4_XS + c0) = oa;
  *(__attribute__((address_space(3))) short8v*)(x_lds + row * V4_XS + c0 + 8) = ob;
}

// NOTE: on amdgpu the 2nd launch_bounds arg is MIN WAVES PER SIMD
// (MI355X_MICROARCH.md): 3 blocks x 512 threads = 6 waves/SIMD, which
// requires the <=80-VGPR allocation bucket.
extern "C" __global__ __launch_bounds__(V4_THREADS, 6)
void dmx_bert_fused_bf16_v4(const unsigned char* __restrict__ lines,
                            const int* __restrict__ start,
                            const int* __restrict__ end,
                            const short* __restrict__ wb,
                            const float* __restrict__ fb,
                            float* __restrict__ scores, int B, int max_len,
                            int n_layers, float eps) {
  const int line = blockIdx.x;
  if (line >= B) return;
  const int tid = threadIdx.x;
  const int wid = tid / DMX_WAVE;
  const int lane = tid % DMX_WAVE;

  extern __shared__ __attribute__((aligned(16))) short smem_raw[];
  v4_lds_short* smem = (v4_lds_short*)smem_raw;
  v4_lds_short* x_lds = smem;                       // [64][136]
  v4_lds_short* hbuf = x_lds + V4_X_ELEMS;          // [64][136] Q|K / P|O / FFN
  v4_lds_short* vt = hbuf + V4_B_ELEMS;             // [64][72] head V^T
  v4_lds_float* smax = (v4_lds_float*)(vt + V4_VT_ELEMS);  // [128]
  v4_lds_float* ssum = smax + 128;                          // [128]

  // ---- embed ----
  {
    const int s0 = start[line], e0 = end[line];
    for (int i = tid * 8; i < V4_S * V4_H; i += V4_THREADS * 8) {
      const int s = i / V4_H, c = i % V4_H;
      int tok = 0;
      const int idx = s0 + s;
      if (idx < e0 && idx < max_len)
        tok = (int)lines[(long)line * max_len + idx] + 3;
      short8v te = *(const short8v*)(wb + WB_TOK + (long)tok * V4_H + c);
      short8v pe = *(const short8v*)(wb + WB_POS + (long)s * V4_H + c);
      short8v xv;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        xv[j] = f32_to_bf16(bf16_to_f32(te[j]) + bf16_to_f32(pe[j]));
      *(__attribute__((address_space(3))) short8v*)(x_lds + s * V4_XS + c) = xv;
    }
  }
  __syncthreads();

  for (int layer = 0; layer < n_layers; ++layer) {
    const short* lw = wb + WB_LAYER0 + (long)layer * LW_SIZE;
    const float* lf = fb + (long)layer * FB_SIZE;

#pragma unroll
    for (int head = 0; head < 2; ++head) {
      v4_qkv_head(x_lds, (v4_glob_cshort*)(lw + LW_QKV),
                  (v4_glob_cfloat*)(lf + FB_BQKV), hbuf, vt, head, wid, lane);
      __syncthreads();
      v4_attention_head(hbuf, vt, smax, ssum, wid, lane);
      __syncthreads();
      // proj partial: x += O_h @ Wo[:, head*64 : head*64+64]
      v4_gemm128<V4_DH, 2, 0, V4_H>(
          hbuf + V4_DH, V4_BS,
          (v4_glob_cshort*)(lw + LW_WO + head * V4_DH),
          head == 0 ? (v4_glob_cfloat*)(lf + FB_BO) : nullptr, nullptr, 0,
          x_lds, wid, lane);
      __syncthreads();
    }
    v4_layernorm(x_lds, lw + LW_LN1G, lw + LW_LN1B, wid, lane, eps);
    __syncthreads();

#pragma unroll
    for (int q = 0; q < 4; ++q) {
      v4_gemm128<V4_H, 0, 1, V4_H>(
          x_lds, V4_XS,
          (v4_glob_cshort*)(lw + LW_W1 + (long)q * (V4_FFN / 4) * V4_H),
          (v4_glob_cfloat*)(lf + FB_B1 + q * (V4_FFN / 4)), hbuf, V4_BS,
          nullptr, wid, lane);
      __syncthreads();
      v4_gemm128<V4_FFN / 4, 2, 0, V4_FFN>(
          hbuf, V4_BS, (v4_glob_cshort*)(lw + LW_W2 + q * (V4_FFN / 4)),
          q == 0 ? (v4_glob_cfloat*)(lf + FB_B2) : nullptr, nullptr, 0, x_lds,
          wid, lane);
      __syncthreads();
    }
    v4_layernorm(x_lds, lw + LW_LN2G, lw + LW_LN2B, wid, lane, eps);
    __syncthreads();
  }

  // ---- pool + score (red aliases smax/ssum: attention is done) ----
  {
    v4_lds_float* red = smax;  // 128 floats
    if (tid < V4_H) {
      float s = 0.f;
      for (int row = 0; row < V4_S; ++row)
        s += bf16_to_f32(x_lds[row * V4_XS + tid]);
      const float w = bf16_to_f32(wb[WB_LAYER0 + (long)n_layers * LW_SIZE + tid]);
      red[tid] = (s / V4_S) * w;
    }
    __syncthreads();
    if (wid == 0) {
      float v = red[lane] + red[lane + 64];
      v = warp_reduce_sum_f32(v);
      if (lane == 0)
        scores[line] = v + fb[(long)n_layers * FB_SIZE];
    }
  }
}

extern "C" void dmx_launch_bert_fused_bf16_v4(
    const void* lines, const void* start, const void* end, const void* wb,
    const void* fb, void* scores, int B, int max_len, int n_layers, float eps,
    hipStream_t stream) {
  const size_t lds =
      (size_t)(V4_X_ELEMS + V4_B_ELEMS + V4_VT_ELEMS) * sizeof(short) +
      V4_F32_ELEMS * sizeof(float);
  static bool attr_set = false;
  if (!attr_set) {
    hipFuncSetAttribute((const void*)dmx_bert_fused_bf16_v4,
                        hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);
    attr_set = true;
  }
  hipLaunchKernelGGL(dmx_bert_fused_bf16_v4, dim3(B), dim3(V4_THREADS), lds,
                     stream, (const unsigned char*)lines, (const int*)start,
                     (const int*)end, (const short*)wb, (const float*)fb,
                     (float*)scores, B, max_len, n_layers, eps);
}
