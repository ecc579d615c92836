// GPU hash-set probe/insert for the NewValue detector family.
//
// The reference's NewValueDetector keeps per-watched-field Python sets
// (docs/getting_started.md:421-511). Here the known-value
// sets are open-addressing u64 hash tables resident in HBM; a batch of
// parsed lines is scored with one kernel launch: for each (line, watch)
// the watched capture span is located, FNV-1a-64 hashed from the line
// bytes, and probed (detect) or inserted (train).
//
// Tables: [W, capacity] u64, 0 = empty (hashes are forced odd). Linear
// probing; capacity is a power of two sized >= 4x expected cardinality.
#include "common.h"

#define EMPTY_KEY 0ull

static __device__ __forceinline__ unsigned long long fnv1a64(
    const unsigned char* p, int n, int lower) {
  unsigned long long h = 1469598103934665603ull;
  for (int i = 0; i < n; ++i) {
    unsigned char c = p[i];
    if (lower && c >= 'A' && c <= 'Z') c += 32;
    h ^= (unsigned long long)c;
    h *= 1099511628211ull;
  }
  return h | 1ull;  // never EMPTY_KEY
}

// Watch spec (mirrors library/detectors/new_value.py::_WatchSpec):
//   kind 0 = content variable: capture index `pos` of lines whose
//            event_id == event (event < 0 -> any event)  [caps table]
//   kind 1 = header variable: format capture index `pos`  [fmt_caps table]
struct WatchSpec {
  int kind;
  int event;
  int pos;
  int _pad;
};

// For each (line, watch): compute the hash of the watched span (0 if the
// field is absent for that line).
extern "C" __global__ __launch_bounds__(256)
void dmx_watch_hashes(
    const unsigned char* __restrict__ lines, int max_len,
    const int* __restrict__ event_id,
    const int* __restrict__ caps, const int* __restrict__ n_caps, int max_caps,
    const int* __restrict__ fmt_caps, const int* __restrict__ n_fmt_caps,
    int max_fmt_caps,
    const WatchSpec* __restrict__ specs, int W,
    int B, int lower,
    unsigned long long* __restrict__ hashes) {  // [B, W]
  const int idx = blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (long)B * W) return;
  const int line = idx / W;
  const int w = idx % W;
  const WatchSpec s = specs[w];
  unsigned long long h = 0ull;

  int start = -1, end = -1;
  if (s.kind == 0) {
    if ((s.event < 0 || event_id[line] == s.event) && s.pos < n_caps[line]) {
      start = caps[((long)line * max_caps + s.pos) * 2];
      end = caps[((long)line * max_caps + s.pos) * 2 + 1];
    }
  } else {
    if (n_fmt_caps && s.pos < n_fmt_caps[line]) {
      start = fmt_caps[((long)line * max_fmt_caps + s.pos) * 2];
      end = fmt_caps[((long)line * max_fmt_caps + s.pos) * 2 + 1];
    }
  }
  if (start >= 0 && end >= start)
    h = fnv1a64(lines + (long)line * max_len + start, end - start, lower);
  hashes[idx] = h;
}

extern "C" __global__ __launch_bounds__(256)
void dmx_hashset_insert(
    const unsigned long long* __restrict__ hashes,  // [B, W]
    unsigned long long* __restrict__ tables,        // [W, capacity]
    int B, int W, int capacity) {
  const int idx = blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (long)B * W) return;
  const unsigned long long h = hashes[idx];
  if (h == EMPTY_KEY) return;
  const int w = idx % W;
  unsigned long long* table = tables + (long)w * capacity;
  unsigned int slot = (unsigned int)(h >> 32 ^ h) & (capacity - 1);
  for (int probe = 0; probe < capacity; ++probe) {
    unsigned long long prev = atomicCAS(
        (unsigned long long*)&table[slot], EMPTY_KEY, h);
    if (prev == EMPTY_KEY || prev == h) return;
    slot = (slot + 1) & (capacity - 1);
  }
  // table full: drop (host sizes capacity with 4x headroom and monitors fill)
}

extern "C" __global__ __launch_bounds__(256)
void dmx_hashset_probe(
    const unsigned long long* __restrict__ hashes,  // [B, W]
    const unsigned long long* __restrict__ tables,  // [W, capacity]
    int B, int W, int capacity,
    int* __restrict__ unseen) {  // [B, W]: 1 = value present but never seen
  const int idx = blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (long)B * W) return;
  const unsigned long long h = hashes[idx];
  int flag = 0;
  if (h != EMPTY_KEY) {
    const int w = idx % W;
    const unsigned long long* table = tables + (long)w * capacity;
    unsigned int slot = (unsigned int)(h >> 32 ^ h) & (capacity - 1);
    flag = 1;
    for (int probe = 0; probe < capacity; ++probe) {
      const unsigned long long v = table[slot];
      if (v == h) { flag = 0; break; }      // known
      if (v == EMPTY_KEY) break;            // definitely unseen
      slot = (slot + 1) & (capacity - 1);
    }
  }
  unseen[idx] = flag;
}

extern "C" void dmx_launch_watch_hashes(
    const void* lines, int max_len, const void* event_id, const void* caps,
    const void* n_caps, int max_caps, const void* fmt_caps,
    const void* n_fmt_caps, int max_fmt_caps, const void* specs, int W,
    int B, int lower, void* hashes, hipStream_t stream) {
  const long total = (long)B * W;
  const int grid = (int)((total + 255) / 256);
  hipLaunchKernelGGL(dmx_watch_hashes, dim3(grid), dim3(256), 0, stream,
                     (const unsigned char*)lines, max_len,
                     (const int*)event_id, (const int*)caps,
                     (const int*)n_caps, max_caps, (const int*)fmt_caps,
                     (const int*)n_fmt_caps, max_fmt_caps,
                     (const WatchSpec*)specs, W, B, lower,
                     (unsigned long long*)hashes);
}

extern "C" void dmx_launch_hashset_insert(
    const void* hashes, void* tables, int B, int W, int capacity,
    hipStream_t stream) {
  const long total = (long)B * W;
  const int grid = (int)((total + 255) / 256);
  hipLaunchKernelGGL(dmx_hashset_insert, dim3(grid), dim3(256), 0, stream,
                     (const unsigned long long*)hashes,
                     (unsigned long long*)tables, B, W, capacity);
}

extern "C" void dmx_launch_hashset_probe(
    const void* hashes, const void* tables, int B, int W, int capacity,
    void* unseen, hipStream_t stream) {
  const long total = (long)B * W;
  const int grid = (int)((total + 255) / 256);
  hipLaunchKernelGGL(dmx_hashset_probe, dim3(grid), dim3(256), 0, stream,
                     (const unsigned long long*)hashes,
                     (const unsigned long long*)tables, B, W, capacity,
                     (int*)unseen);
}
