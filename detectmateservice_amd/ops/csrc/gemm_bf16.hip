// Fused linear layer for MI355X (gfx950): C = act(A @ W + bias), bf16.
//
// MI355X-first design (guide: cdna_hip_programming.md §5 anatomy):
//  * A is [M, K] row-major activations; the weight is pre-transposed on the
//    host to Wt [N, K] row-major so BOTH MFMA operands are K-contiguous
//    (16-B ds_read_b128 fragments, no transpose reads needed).
//  * mfma_f32_16x16x32_bf16 per-wave tiles, 4 waves per block, 128x128
//    block tile, BK=64, double-buffered LDS staged with
//    __builtin_amdgcn_global_load_lds width 16 (async global->LDS DMA).
//  * XOR swizzle on the glds SOURCE address + the matching XOR on the
//    ds_read offset (guide §5.4 rule 21: linear LDS dest, inverse-swizzled
//    source): rows of a tile are 128 B, read by 16-lane groups at a fixed
//    16-B unit -> up to 8-way bank conflict unswizzled; unit ^= (row&7)
//    spreads the group over all 8 units of the row.
//  * XCD-aware bijective blockIdx swizzle (guide §5.5 T1) for L2 affinity
//    across the 8 XCDs.
//
// Replaces (capability-wise) the detector scoring matmuls the reference
// delegates to sklearn/numpy in `detectmatelibrary`.

#include "common.h"

#define BM 128
#define BN 128
#define BK 64
#define N_WAVES 4
#define THREADS (N_WAVES * DMX_WAVE)

// Epilogue codes
#define EPI_NONE 0
#define EPI_GELU 1
#define EPI_RELU 2

typedef __attribute__((address_space(3))) void lds_void;
typedef const __attribute__((address_space(1))) void global_void;

// Stage one [rows=128][BK=64] bf16 tile (16 KiB) from global (row stride
// `ld_elems` bf16) into LDS, lane-linear destination, source XOR-swizzled.
// rows beyond `max_row` are clamped (duplicate loads, never OOB).
static __device__ __forceinline__ void stage_tile_glds(
    const short* __restrict__ gbase,  // element pointer at (row0, k0)
    int ld_elems, int max_row_excl,   // rows available from row0
    short* lds_tile) {
  const int t = threadIdx.x;          // 0..255
  const int wid = t / DMX_WAVE;       // 4 waves: each stages 4x 1 KiB chunks
  const int lane = t % DMX_WAVE;
#pragma unroll
  for (int chunk = 0; chunk < 4; ++chunk) {
    // linear 16-B unit index this thread fills
    const int unit_lin = chunk * 256 + t;
    const int row = unit_lin >> 3;                   // 8 units per 128-B row
    const int unit = unit_lin & 7;
    const int src_unit = unit ^ (row & 7);           // inverse swizzle
    int src_row = row < max_row_excl ? row : (max_row_excl - 1);
    const short* src = gbase + (long)src_row * ld_elems + src_unit * 8;
    // LDS dest: wave-uniform base; lane-linear 16 B per lane.
    short* dst = lds_tile + (chunk * 4096 + wid * 1024) / 2;
    __builtin_amdgcn_global_load_lds(
        (global_void*)src, (lds_void*)dst, 16, 0, 0);
  }
}

// Read an 8-element bf16 MFMA fragment from a staged tile.
// row = frag_row + (lane&15); k units: (lane>>4) within a 32-wide k-step.
static __device__ __forceinline__ bf16x8 read_frag(
    const short* lds_tile, int frag_row, int kstep, int lane) {
  const int row = frag_row + (lane & 15);
  const int unit = (kstep * 4 + (lane >> 4)) ^ (row & 7);
  const short* p = lds_tile + row * 64 + unit * 8;
  return *(const bf16x8*)p;
}

extern "C" __global__ __launch_bounds__(THREADS)
void dmx_fused_linear_bf16(
    const short* __restrict__ A,   // [M, K] bf16
    const short* __restrict__ Wt,  // [N, K] bf16 (W transposed)
    const float* __restrict__ bias,  // [N] or nullptr
    short* __restrict__ C,         // [M, N] bf16
    int M, int N, int K, int epilogue, int grid_m) {
  // XCD-aware bijective block swizzle (T1): contiguous grid chunks per XCD.
  const int nwg = gridDim.x;
  int bid = blockIdx.x;
  {
    const int nx = 8;
    const int q = nwg / nx, r = nwg % nx;
    const int xcd = bid % nx, idx = bid / nx;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int bm = bid % grid_m;
  const int bn = bid / grid_m;
  const int m0 = bm * BM;
  const int n0 = bn * BN;

  const int t = threadIdx.x;
  const int wid = t / DMX_WAVE;
  const int lane = t % DMX_WAVE;
  const int wave_m = (wid >> 1) * 64;  // 2x2 wave grid, 64x64 per wave
  const int wave_n = (wid & 1) * 64;

  // LDS: double-buffered A and B tiles; ONE __shared__ object (guide §5
  // .s-level trap (a): a second __shared__ forces vmcnt(0) per ds_read).
  __shared__ __attribute__((aligned(16))) short lds[2 * 2 * BM * BK];
  // buffer layout: [buf][A|B][BM*BK]; pointer computed per use (an array of
  // LDS pointers is not a valid static initializer under hipcc)
#define LDS_A(buf) (lds + (buf) * 2 * BM * BK)
#define LDS_B(buf) (lds + (buf) * 2 * BM * BK + BM * BK)

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int n_tiles = K / BK;  // host asserts K % 64 == 0
  int cur = 0;
  stage_tile_glds(A + (long)m0 * K, K, M - m0, LDS_A(0));
  stage_tile_glds(Wt + (long)n0 * K, K, N - n0, LDS_B(0));
  __syncthreads();  // drains glds (vmcnt 0) + barrier

  for (int kt = 0; kt < n_tiles; ++kt) {
    if (kt + 1 < n_tiles) {
      stage_tile_glds(A + (long)m0 * K + (kt + 1) * BK, K, M - m0, LDS_A(cur ^ 1));
      stage_tile_glds(Wt + (long)n0 * K + (kt + 1) * BK, K, N - n0, LDS_B(cur ^ 1));
    }
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {  // two 32-wide k-steps per BK=64
      bf16x8 a_frag[4], b_frag[4];
#pragma unroll
      for (int f = 0; f < 4; ++f)
        a_frag[f] = read_frag(LDS_A(cur), wave_m + f * 16, ks, lane);
#pragma unroll
      for (int f = 0; f < 4; ++f)
        b_frag[f] = read_frag(LDS_B(cur), wave_n + f * 16, ks, lane);
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 4; ++fn)
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[fm], b_frag[fn], acc[fm][fn], 0, 0, 0);
    }
    __syncthreads();
    cur ^= 1;
  }

  // Epilogue: bias + activation + bf16 store.
  // C/D mapping (guide §3): col = lane&15, row = (lane>>4)*4 + reg.
  const int col_in_frag = lane & 15;
  const int row_base = (lane >> 4) * 4;
#pragma unroll
  for (int fm = 0; fm < 4; ++fm) {
#pragma unroll
    for (int fn = 0; fn < 4; ++fn) {
      const int n = n0 + wave_n + fn * 16 + col_in_frag;
      if (n >= N) continue;
      const float b = bias ? bias[n] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = m0 + wave_m + fm * 16 + row_base + r;
        if (m >= M) continue;
        float v = acc[fm][fn][r] + b;
        if (epilogue == EPI_GELU) v = gelu_f32(v);
        else if (epilogue == EPI_RELU) v = fmaxf(v, 0.f);
        C[(long)m * N + n] = f32_to_bf16(v);
      }
    }
  }
}

extern "C" void dmx_launch_fused_linear_bf16(
    const void* A, const void* Wt, const void* bias, void* C,
    int M, int N, int K, int epilogue, hipStream_t stream) {
  const int grid_m = (M + BM - 1) / BM;
  const int grid_n = (N + BN - 1) / BN;
  dim3 grid(grid_m * grid_n);
  dim3 block(THREADS);
  hipLaunchKernelGGL(dmx_fused_linear_bf16, grid, block, 0, stream,
                     (const short*)A, (const short*)Wt, (const float*)bias,
                     (short*)C, M, N, K, epilogue, grid_m);
}

// ---------------------------------------------------------------------------
// MFMA layout probe: one wave computes D = A(16x32) @ B(32x16) with a
// selectable operand-fragment layout; the GPU test (tests/test_gpu_ops.py)
// verifies layout 0 against a torch reference and the C/D mapping.
// layout 0: lane l holds X[l&15][(l>>4)*8 + j]   (contiguous k octet)
// layout 1: lane l holds X[l&15][(l>>4)*4 + (j&3) + (j>>2)*16]  (split octet)
// ---------------------------------------------------------------------------
extern "C" __global__ void dmx_probe_mfma_16x16x32(
    const short* __restrict__ A,  // [16][32] bf16 row-major
    const short* __restrict__ B,  // [32][16] bf16 row-major (k-major rows)
    float* __restrict__ D,        // [16][16] f32 row-major
    int a_layout, int b_layout) {
  const int lane = threadIdx.x & 63;
  bf16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int ka = a_layout == 0 ? (lane >> 4) * 8 + j
                           : (lane >> 4) * 4 + (j & 3) + (j >> 2) * 16;
    a[j] = A[(lane & 15) * 32 + ka];
    int kb = b_layout == 0 ? (lane >> 4) * 8 + j
                           : (lane >> 4) * 4 + (j & 3) + (j >> 2) * 16;
    b[j] = B[kb * 16 + (lane & 15)];
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = (lane >> 4) * 4 + r;
    const int col = lane & 15;
    D[row * 16 + col] = acc[r];
  }
}

extern "C" void dmx_launch_probe_mfma(
    const void* A, const void* B, void* D, int a_layout, int b_layout,
    hipStream_t stream) {
  hipLaunchKernelGGL(dmx_probe_mfma_16x16x32, dim3(1), dim3(64), 0, stream,
                     (const short*)A, (const short*)B, (float*)D, a_layout,
                     b_layout);
}

// ---------------------------------------------------------------------------
// Layout probe for mfma_f32_32x32x16_bf16 (A 32x16, B 16x32, D 32x32).
// Assumed (verified empirically on gfx950 by tests/test_gpu_ops.py):
//   A: lane l holds A[l&31][(l>>5)*8 + j], j = 0..7 (contiguous k-octet)
//   B: lane l holds B[(l>>5)*8 + j][l&31]
//   C/D (guide §3): col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
// ---------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(16))) float f32x16;

extern "C" __global__ void dmx_probe_mfma_32x32x16(
    const short* __restrict__ A,  // [32][16] bf16
    const short* __restrict__ B,  // [16][32] bf16
    float* __restrict__ D) {      // [32][32] f32
  const int lane = threadIdx.x & 63;
  bf16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = A[(lane & 31) * 16 + (lane >> 5) * 8 + j];
    b[j] = B[((lane >> 5) * 8 + j) * 32 + (lane & 31)];
  }
  f32x16 acc;
#pragma unroll
  for (int i = 0; i < 16; ++i) acc[i] = 0.f;
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
    const int col = lane & 31;
    D[row * 32 + col] = acc[r];
  }
}

extern "C" void dmx_launch_probe_mfma32(const void* A, const void* B, void* D,
                                        hipStream_t stream) {
  hipLaunchKernelGGL(dmx_probe_mfma_32x32x16, dim3(1), dim3(64), 0, stream,
                     (const short*)A, (const short*)B, (float*)D);
}
