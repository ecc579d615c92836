// Shared LogSchema proto3 decode core — pure C++ (no Python API), safe to
// run with the GIL released. Used by codec.cpp (batch decode entry points)
// and frame_reader.cpp (socket -> packed tensors fast path).
#pragma once

#include <cstdint>
#include <cstring>
#include <utility>
#include <vector>

namespace dmx_proto {

inline bool get_varint(const uint8_t* p, size_t n, size_t& pos, uint64_t& v) {
  v = 0;
  int shift = 0;
  while (pos < n && shift < 70) {
    uint8_t b = p[pos++];
    v |= (uint64_t)(b & 0x7F) << shift;
    if (!(b & 0x80)) return true;
    shift += 7;
  }
  return false;
}

// All length arithmetic below is overflow-safe: get_varint leaves pos <= n,
// so `len > n - pos` rejects any oversized length without computing pos+len
// (a 10-byte crafted varint near 2^64 would wrap pos+len below n and turn
// the bounds check into an out-of-bounds read).
inline bool skip_field(const uint8_t* p, size_t n, size_t& pos, int wt) {
  uint64_t tmp;
  switch (wt) {
    case 0: return get_varint(p, n, pos, tmp);
    case 1:
      if (n - pos < 8) return false;
      pos += 8;
      return true;
    case 2:
      if (!get_varint(p, n, pos, tmp) || tmp > (uint64_t)(n - pos)) return false;
      pos += (size_t)tmp;
      return true;
    case 5:
      if (n - pos < 4) return false;
      pos += 4;
      return true;
    default: return false;
  }
}

struct LogSpan {  // byte spans into the source frame (no copies)
  const char* id = "";   size_t id_len = 0;
  const char* src = "";  size_t src_len = 0;
  const char* host = ""; size_t host_len = 0;
};

// varint walk + line copy for one batch of LogSchema frames.
inline void decode_log_core(
    const std::vector<std::pair<const uint8_t*, size_t>>& raw,
    int64_t max_len, uint8_t* lbuf, int32_t* lenp,
    std::vector<LogSpan>& spans) {
  const int64_t B = (int64_t)raw.size();
  for (int64_t i = 0; i < B; ++i) {
    const uint8_t* p = raw[i].first;
    const size_t n = raw[i].second;
    size_t pos = 0;
    LogSpan& sp = spans[i];
    while (pos < n) {
      uint64_t key;
      if (!get_varint(p, n, pos, key)) break;
      const int field = (int)(key >> 3), wt = (int)(key & 7);
      if (wt == 2) {
        uint64_t sl;
        if (!get_varint(p, n, pos, sl) || sl > (uint64_t)(n - pos)) break;
        const char* s = (const char*)(p + pos);
        switch (field) {
          case 2: sp.id = s; sp.id_len = sl; break;
          case 3: {  // log line -> packed buffer
            const size_t copy = sl < (uint64_t)max_len ? sl : (size_t)max_len;
            std::memcpy(lbuf + i * max_len, s, copy);
            lenp[i] = (int32_t)copy;
            break;
          }
          case 4: sp.src = s; sp.src_len = sl; break;
          case 5: sp.host = s; sp.host_len = sl; break;
          default: break;
        }
        pos += sl;
      } else {
        if (!skip_field(p, n, pos, wt)) break;
      }
    }
  }
}

}  // namespace dmx_proto
