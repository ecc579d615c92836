// Fused residual-add + LayerNorm, bf16 in/out, f32 accumulation.
//
// Memory-bound streaming op: one wave per row, vectorized bf16x8 loads
// (guide Guideline 13: scalar bf16 loads cost ~2x), f32 wave-shuffle
// reductions, fused residual add so the row is read once (HBM3E-bound
// ops get fused into the producing pass — SURVEY.md build mandate).
#include "common.h"

extern "C" __global__ __launch_bounds__(256)
void dmx_layernorm_bf16(
    const short* __restrict__ X,     // [M, D] bf16
    const short* __restrict__ R,     // [M, D] bf16 residual or nullptr
    const short* __restrict__ gamma, // [D] bf16
    const short* __restrict__ beta,  // [D] bf16
    short* __restrict__ Y,           // [M, D] bf16
    short* __restrict__ Xres,        // [M, D] bf16: x+res (for next residual) or nullptr
    int M, int D, float eps) {
  const int wave = (blockIdx.x * (256 / DMX_WAVE)) + threadIdx.x / DMX_WAVE;
  const int lane = threadIdx.x % DMX_WAVE;
  if (wave >= M) return;
  const long row_off = (long)wave * D;

  // Each lane covers D/64 elements; vectorize by 8 when possible.
  const int epl = D / DMX_WAVE;  // host asserts D % 64 == 0
  float vals[32];                // supports D up to 2048
  float sum = 0.f;

  if ((epl & 7) == 0) {
    for (int base = 0; base < epl; base += 8) {
      const int idx = lane * epl + base;  // contiguous 8 per lane
      short8v v = *(const short8v*)(X + row_off + idx);
      short8v rv;
      if (R) rv = *(const short8v*)(R + row_off + idx);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf16_to_f32(v[j]);
        if (R) f += bf16_to_f32(rv[j]);
        vals[base + j] = f;
        sum += f;
      }
    }
  } else {
    for (int e = 0; e < epl; ++e) {
      const int idx = e * DMX_WAVE + lane;  // strided, coalesced
      float f = bf16_to_f32(X[row_off + idx]);
      if (R) f += bf16_to_f32(R[row_off + idx]);
      vals[e] = f;
      sum += f;
    }
  }

  sum = warp_reduce_sum_f32(sum);
  const float mean = __shfl(sum, 0, 64) / D;
  float var = 0.f;
  for (int e = 0; e < epl; ++e) {
    const float d = vals[e] - mean;
    var += d * d;
  }
  var = warp_reduce_sum_f32(var);
  const float rstd = rsqrtf(__shfl(var, 0, 64) / D + eps);

  if ((epl & 7) == 0) {
    for (int base = 0; base < epl; base += 8) {
      const int idx = lane * epl + base;
      short8v g = *(const short8v*)(gamma + idx);
      short8v b = *(const short8v*)(beta + idx);
      short8v out, xr;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float f = vals[base + j];
        out[j] = f32_to_bf16((f - mean) * rstd * bf16_to_f32(g[j]) +
                             bf16_to_f32(b[j]));
        xr[j] = f32_to_bf16(f);
      }
      *(short8v*)(Y + row_off + idx) = out;
      if (Xres) *(short8v*)(Xres + row_off + idx) = xr;
    }
  } else {
    for (int e = 0; e < epl; ++e) {
      const int idx = e * DMX_WAVE + lane;
      const float f = vals[e];
      Y[row_off + idx] = f32_to_bf16(
          (f - mean) * rstd * bf16_to_f32(gamma[idx]) + bf16_to_f32(beta[idx]));
      if (Xres) Xres[row_off + idx] = f32_to_bf16(f);
    }
  }
}

// Row-group variant: LANES_PER_ROW = D/8 lanes per row, each lane loads ONE
// short8 (16 B) — full vector loads at any D in {64,128,256,512}. The
// wave-per-row kernel above falls back for other D. At D=128 this packs 4
// rows per wave (the bench's hidden size; the scalar path measured 13% of
// step time — profiles/r01_bench_b32768_kernel_stats.txt).
template <int LPR>  // lanes per row (power of two, <= 64)
__global__ __launch_bounds__(256) void dmx_layernorm_rowgroup_bf16(
    const short* __restrict__ X, const short* __restrict__ R,
    const short* __restrict__ gamma, const short* __restrict__ beta,
    short* __restrict__ Y, short* __restrict__ Xres, int M, int D,
    float eps) {
  const int rows_per_block = 256 / LPR;
  const int row = blockIdx.x * rows_per_block + threadIdx.x / LPR;
  if (row >= M) return;
  const int sub = threadIdx.x % LPR;       // lane-in-row
  const long off = (long)row * D + sub * 8;

  short8v v = *(const short8v*)(X + off);
  float vals[8];
  float sum = 0.f;
  if (R) {
    short8v rv = *(const short8v*)(R + off);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      vals[j] = bf16_to_f32(v[j]) + bf16_to_f32(rv[j]);
      sum += vals[j];
    }
  } else {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      vals[j] = bf16_to_f32(v[j]);
      sum += vals[j];
    }
  }
#pragma unroll
  for (int mask = 1; mask < LPR; mask <<= 1) sum += __shfl_xor(sum, mask, 64);
  const float mean = sum / D;
  float var = 0.f;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const float d = vals[j] - mean;
    var += d * d;
  }
#pragma unroll
  for (int mask = 1; mask < LPR; mask <<= 1) var += __shfl_xor(var, mask, 64);
  const float rstd = rsqrtf(var / D + eps);

  short8v g = *(const short8v*)(gamma + sub * 8);
  short8v bt = *(const short8v*)(beta + sub * 8);
  short8v out, xr;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    out[j] = f32_to_bf16((vals[j] - mean) * rstd * bf16_to_f32(g[j]) +
                         bf16_to_f32(bt[j]));
    xr[j] = f32_to_bf16(vals[j]);
  }
  *(short8v*)(Y + off) = out;
  if (Xres) *(short8v*)(Xres + off) = xr;
}

extern "C" void dmx_launch_layernorm_bf16(
    const void* X, const void* R, const void* gamma, const void* beta,
    void* Y, void* Xres, int M, int D, float eps, hipStream_t stream) {
  // Row-group fast path for D in {64, 128, 256, 512}.
  const int lpr = D / 8;
  if (D == 64 || D == 128 || D == 256 || D == 512) {
    const int rows_per_block = 256 / lpr;
    const int grid = (M + rows_per_block - 1) / rows_per_block;
    switch (D) {
      case 64:
        hipLaunchKernelGGL(dmx_layernorm_rowgroup_bf16<8>, dim3(grid),
                           dim3(256), 0, stream, (const short*)X,
                           (const short*)R, (const short*)gamma,
                           (const short*)beta, (short*)Y, (short*)Xres, M, D,
                           eps);
        return;
      case 128:
        hipLaunchKernelGGL(dmx_layernorm_rowgroup_bf16<16>, dim3(grid),
                           dim3(256), 0, stream, (const short*)X,
                           (const short*)R, (const short*)gamma,
                           (const short*)beta, (short*)Y, (short*)Xres, M, D,
                           eps);
        return;
      case 256:
        hipLaunchKernelGGL(dmx_layernorm_rowgroup_bf16<32>, dim3(grid),
                           dim3(256), 0, stream, (const short*)X,
                           (const short*)R, (const short*)gamma,
                           (const short*)beta, (short*)Y, (short*)Xres, M, D,
                           eps);
        return;
      case 512:
        hipLaunchKernelGGL(dmx_layernorm_rowgroup_bf16<64>, dim3(grid),
                           dim3(256), 0, stream, (const short*)X,
                           (const short*)R, (const short*)gamma,
                           (const short*)beta, (short*)Y, (short*)Xres, M, D,
                           eps);
        return;
    }
  }
  const int waves_per_block = 256 / DMX_WAVE;
  const int grid = (M + waves_per_block - 1) / waves_per_block;
  hipLaunchKernelGGL(dmx_layernorm_bf16, dim3(grid), dim3(256), 0, stream,
                     (const short*)X, (const short*)R, (const short*)gamma,
                     (const short*)beta, (short*)Y, (short*)Xres, M, D, eps);
}
