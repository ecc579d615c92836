// Batched log parsing: log_format header split + <*>-wildcard template
// matching, one wave per line.
//
// This is the GPU replacement for the per-message Python MatcherParser
// (reference capability: parsers.template_matcher) and for
// the Levenshtein/template native ext the reference library family uses
// (detectmateperformance). The matching algorithm is the
// exact greedy-anchored wildcard match of
// detectmateservice_amd/library/parsers/template_matcher.py::match_template
// (the Python implementation is the semantic reference; parity is tested).
//
// Layout: lines live in a padded SoA byte buffer [B, max_len] (decoded from
// protobuf frames once per batch, SURVEY.md §2.4 "GPU-resident columnar
// batch layout"). Each line is staged into LDS by its wave; the template
// segment blob is staged once per workgroup. Substring search is
// wave-parallel: 64 candidate start positions per step, first match via
// __ballot (64-bit on CDNA) + ffs.
#include "common.h"

#define TM_WAVES 4
#define TM_MAX_LINE 512

// Wave-parallel find of seg[0..sl) in line_lds[pos..limit): returns first
// index or -1; uniform across the wave (ballot is wave-uniform).
static __device__ __forceinline__ int wave_find(
    const unsigned char* line_lds, int pos, int limit,
    const unsigned char* seg, int sl, int lane, int lower) {
  const int last = limit - sl;
  for (int base = pos; base <= last; base += DMX_WAVE) {
    const int cand = base + lane;
    bool ok = cand <= last;
    if (ok) {
      for (int b = 0; b < sl; ++b) {
        unsigned char c = line_lds[cand + b];
        if (lower && c >= 'A' && c <= 'Z') c += 32;
        if (c != seg[b]) { ok = false; break; }
      }
    }
    const unsigned long long mask = __ballot(ok);
    if (mask) return base + (__ffsll((long long)mask) - 1);
  }
  return -1;
}

// Greedy-anchored wildcard match (ports match_template exactly).
// Returns capture count (written to caps as start,end pairs by lane 0)
// or -1 on no-match.
static __device__ int match_segments(
    const unsigned char* line_lds, int start, int end,
    const unsigned char* seg_bytes, const int* seg_off,
    int s_begin, int s_end, int* caps, int max_caps, int lane, int lower) {
  int pos = start;
  int ncap = 0;
  const int nseg = s_end - s_begin;
  for (int i = 0; i < nseg; ++i) {
    const int so = seg_off[s_begin + i];
    const int sl = seg_off[s_begin + i + 1] - so;
    if (sl == 0) {
      if (i == nseg - 1) {
        if (ncap < max_caps && lane == 0) {
          caps[ncap * 2] = pos;
          caps[ncap * 2 + 1] = end;
        }
        return ncap + 1;
      }
      continue;
    }
    const int idx = wave_find(line_lds, pos, end, seg_bytes + so, sl, lane, lower);
    if (idx < 0) return -1;
    if (i == 0 && idx != start) return -1;
    if (i > 0) {
      if (ncap < max_caps && lane == 0) {
        caps[ncap * 2] = pos;
        caps[ncap * 2 + 1] = idx;
      }
      ++ncap;
    }
    pos = idx + sl;
  }
  if (pos != end) return -1;
  return ncap;
}

extern "C" __global__ __launch_bounds__(TM_WAVES * DMX_WAVE)
void dmx_template_match(
    const unsigned char* __restrict__ lines,  // [B, max_len]
    const int* __restrict__ line_len,         // [B]
    int B, int max_len,
    // format "template" (header split): nf_seg == 0 -> no format
    const unsigned char* __restrict__ fmt_bytes,
    const int* __restrict__ fmt_seg_off, int nf_seg,
    // content templates
    const unsigned char* __restrict__ seg_bytes, int seg_bytes_len,
    const int* __restrict__ seg_off,
    const int* __restrict__ tpl_seg_start, int n_tpl,
    int lower,
    // outputs
    int* __restrict__ event_id,     // [B] 1-based or -1
    int* __restrict__ fmt_caps,     // [B, max_fmt_caps, 2]
    int* __restrict__ n_fmt_caps,   // [B]
    int* __restrict__ caps,         // [B, max_caps, 2]
    int* __restrict__ n_caps,       // [B]
    // content span (what the transformer scorer consumes): the last
    // header capture when the format matched, else [0, len) — computed
    // here so the hot loop needs no gather/where chain of ~6 torch
    // kernels per batch (profiles/r10)
    int* __restrict__ span_start,   // [B]
    int* __restrict__ span_end,     // [B]
    int max_fmt_caps, int max_caps) {
  const int wid = threadIdx.x / DMX_WAVE;
  const int lane = threadIdx.x % DMX_WAVE;
  const int line_idx = blockIdx.x * TM_WAVES + wid;

  extern __shared__ __attribute__((aligned(16))) unsigned char smem[];
  unsigned char* seg_lds = smem;                       // [seg_bytes_len]
  unsigned char* line_lds = smem + ((seg_bytes_len + 15) & ~15)
                            + wid * TM_MAX_LINE;       // per-wave line

  // stage segment blob once per block (all threads cooperate)
  for (int i = threadIdx.x; i < seg_bytes_len; i += TM_WAVES * DMX_WAVE)
    seg_lds[i] = seg_bytes[i];
  __syncthreads();

  if (line_idx >= B) return;
  const int len = min(line_len[line_idx], max_len);
  const unsigned char* gline = lines + (long)line_idx * max_len;
  for (int i = lane; i < len; i += DMX_WAVE) line_lds[i] = gline[i];
  // The line staging is per-wave (waves may exit divergently, so no
  // __syncthreads here): drain the wave's own LDS writes before cross-lane
  // reads. asm form with "memory" clobber so the compiler cannot hoist the
  // following LDS reads above it (guide §5.4 rule 18 hazard).
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

  int content_start = 0, content_end = len;
  int nfc = 0;
  if (nf_seg > 0) {
    nfc = match_segments(line_lds, 0, len, fmt_bytes, fmt_seg_off, 0, nf_seg,
                         fmt_caps + (long)line_idx * max_fmt_caps * 2,
                         max_fmt_caps, lane, 0);
    if (nfc > 0) {
      // content = LAST capture (the trailing token convention of
      // MatcherParser._content_field)
      const int last = min(nfc, max_fmt_caps) - 1;
      if (lane == 0) {
        content_start = fmt_caps[(long)line_idx * max_fmt_caps * 2 + last * 2];
        content_end = fmt_caps[(long)line_idx * max_fmt_caps * 2 + last * 2 + 1];
      }
      content_start = __shfl(content_start, 0, 64);
      content_end = __shfl(content_end, 0, 64);
    } else {
      nfc = 0;  // -1 -> no header
    }
  }
  if (n_fmt_caps) n_fmt_caps[line_idx] = nfc;

  int eid = -1, nc = 0;
  for (int t = 0; t < n_tpl; ++t) {
    const int r = match_segments(
        line_lds, content_start, content_end, seg_lds, seg_off,
        tpl_seg_start[t], tpl_seg_start[t + 1],
        caps + (long)line_idx * max_caps * 2, max_caps, lane, lower);
    if (r >= 0) {
      eid = t + 1;
      nc = r;
      break;
    }
  }
  if (lane == 0) {
    event_id[line_idx] = eid;
    n_caps[line_idx] = nc;
    if (span_start) {
      span_start[line_idx] = content_start;
      span_end[line_idx] = content_end;
    }
  }
}

extern "C" void dmx_launch_template_match(
    const void* lines, const void* line_len, int B, int max_len,
    const void* fmt_bytes, const void* fmt_seg_off, int nf_seg,
    const void* seg_bytes, int seg_bytes_len, const void* seg_off,
    const void* tpl_seg_start, int n_tpl, int lower,
    void* event_id, void* fmt_caps, void* n_fmt_caps, void* caps,
    void* n_caps, void* span_start, void* span_end, int max_fmt_caps,
    int max_caps, hipStream_t stream) {
  const int grid = (B + TM_WAVES - 1) / TM_WAVES;
  const size_t lds = ((seg_bytes_len + 15) & ~15) + TM_WAVES * TM_MAX_LINE;
  hipLaunchKernelGGL(dmx_template_match, dim3(grid), dim3(TM_WAVES * DMX_WAVE),
                     lds, stream,
                     (const unsigned char*)lines, (const int*)line_len, B,
                     max_len, (const unsigned char*)fmt_bytes,
                     (const int*)fmt_seg_off, nf_seg,
                     (const unsigned char*)seg_bytes, seg_bytes_len,
                     (const int*)seg_off, (const int*)tpl_seg_start, n_tpl,
                     lower, (int*)event_id, (int*)fmt_caps, (int*)n_fmt_caps,
                     (int*)caps, (int*)n_caps, (int*)span_start,
                     (int*)span_end, max_fmt_caps, max_caps);
}
