// Batched Levenshtein edit distance: anti-diagonal wavefront DP in LDS.
//
// Replaces (capability-wise) the reference family's native Levenshtein
// extension (`detectmateperformance`, pybind11 + C ext — SURVEY.md §2.4):
// used by template mining (auto_config) to cluster log lines by string
// similarity. One wave per (a, b) pair; both strings staged in LDS; the
// DP runs over anti-diagonals (cells (i,j) with i+j=k depend only on
// diagonals k-1 and k-2, so all cells of a diagonal compute in parallel
// across lanes). Diagonal arrays live in LDS; a wave-local
// s_waitcnt lgkmcnt(0) orders neighbor-lane reads between diagonals.
#include "common.h"

#define ED_WAVES 4
#define ED_MAX_LEN 256

extern "C" __global__ __launch_bounds__(ED_WAVES * DMX_WAVE)
void dmx_edit_distance(
    const unsigned char* __restrict__ A, const int* __restrict__ a_len,
    int Na,
    const unsigned char* __restrict__ B, const int* __restrict__ b_len,
    int Nb, int max_len,
    int* __restrict__ dist) {  // [Na, Nb]
  const int wid = threadIdx.x / DMX_WAVE;
  const int lane = threadIdx.x % DMX_WAVE;
  const long pair = (long)blockIdx.x * ED_WAVES + wid;
  if (pair >= (long)Na * Nb) return;
  const int ia = (int)(pair / Nb);
  const int ib = (int)(pair % Nb);
  const int m = min(a_len[ia], ED_MAX_LEN);  // rows (string a)
  const int n = min(b_len[ib], ED_MAX_LEN);  // cols (string b)

  extern __shared__ __attribute__((aligned(16))) unsigned char smem[];
  // per-wave carve: a[256] + b[256] + 3 diagonal arrays of (ED_MAX_LEN+1) u16
  const int wave_bytes = 2 * ED_MAX_LEN + 3 * (ED_MAX_LEN + 1) * 2 + 16;
  unsigned char* base = smem + wid * wave_bytes;
  unsigned char* a_s = base;
  unsigned char* b_s = base + ED_MAX_LEN;
  unsigned short* diag = (unsigned short*)(base + 2 * ED_MAX_LEN + (16 - (2 * ED_MAX_LEN) % 16) % 16);
  unsigned short* d0 = diag;                       // k-2
  unsigned short* d1 = diag + (ED_MAX_LEN + 1);    // k-1
  unsigned short* d2 = diag + 2 * (ED_MAX_LEN + 1);  // k (current)

  for (int i = lane; i < m; i += DMX_WAVE) a_s[i] = A[(long)ia * max_len + i];
  for (int j = lane; j < n; j += DMX_WAVE) b_s[j] = B[(long)ib * max_len + j];
  // init: d0 = diagonal k=0 { (0,0)=0 }, d1 = diagonal k=1 { (1,0)=1,(0,1)=1 }
  if (lane == 0) {
    d0[0] = 0;
    d1[0] = 1;  // cell (i=1, j=0) stored at index j=0
    d1[1] = 1;  // cell (i=0, j=1)
  }
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

  if (m == 0 || n == 0) {
    if (lane == 0) dist[(long)ia * Nb + ib] = m + n;
    return;
  }

  // diagonal k (2..m+n): cells (i, j) with i + j = k, indexed by j.
  unsigned short result = 0;
  for (int k = 2; k <= m + n; ++k) {
    const int j_lo = max(0, k - m);
    const int j_hi = min(k, n);
    for (int j = j_lo + lane; j <= j_hi; j += DMX_WAVE) {
      const int i = k - j;
      unsigned short v;
      if (j == 0) {
        v = (unsigned short)i;  // first column: distance = i
      } else if (i == 0) {
        v = (unsigned short)j;  // first row: distance = j
      } else {
        // neighbors: (i-1, j) is on diag k-1 at index j;
        //            (i, j-1) is on diag k-1 at index j-1;
        //            (i-1, j-1) is on diag k-2 at index j-1.
        const unsigned short up = d1[j];
        const unsigned short left = d1[j - 1];
        const unsigned short ul = d0[j - 1];
        const unsigned short cost = (a_s[i - 1] == b_s[j - 1]) ? 0 : 1;
        v = min((unsigned short)(min(up, left) + 1),
                (unsigned short)(ul + cost));
      }
      d2[j] = v;
      if (i == m && j == n) result = v;
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    // rotate diagonals
    unsigned short* tmp = d0;
    d0 = d1;
    d1 = d2;
    d2 = tmp;
  }
  // Only the lane that computed cell (m, n) holds the result; every other
  // lane holds 0 and Levenshtein(m,n) == 0 only when the strings are
  // equal (then every lane agrees on 0) -> max-reduce is exact.
  int r = result;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) r = max(r, __shfl_down(r, off, 64));
  r = __shfl(r, 0, 64);
  if (lane == 0) dist[(long)ia * Nb + ib] = r;
}

extern "C" void dmx_launch_edit_distance(
    const void* A, const void* a_len, int Na, const void* B,
    const void* b_len, int Nb, int max_len, void* dist,
    hipStream_t stream) {
  const long pairs = (long)Na * Nb;
  const int grid = (int)((pairs + ED_WAVES - 1) / ED_WAVES);
  const int wave_bytes = 2 * ED_MAX_LEN + 3 * (ED_MAX_LEN + 1) * 2 + 16;
  const size_t lds = (size_t)ED_WAVES * wave_bytes;
  hipLaunchKernelGGL(dmx_edit_distance, dim3(grid), dim3(ED_WAVES * DMX_WAVE),
                     lds, stream, (const unsigned char*)A, (const int*)a_len,
                     Na, (const unsigned char*)B, (const int*)b_len, Nb,
                     max_len, (int*)dist);
}
