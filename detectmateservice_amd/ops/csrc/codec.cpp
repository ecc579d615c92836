// Batched proto3 codec (CPU): N frames <-> SoA buffers in one call.
//
// The reference decodes/encodes one protobuf message per Python call in
// the per-message loop; this framework's engine is
// batch-first, so the codec is too: decode a batch of LogSchema frames
// into the packed [B, max_len] byte buffer the parser kernel consumes,
// and encode a batch of ParserSchema frames from the kernel's span
// outputs without touching per-message Python objects (SURVEY.md §2.4
// "GPU-resident columnar batch layout").
//
// Wire format identical to detectmateservice_amd/schemas/codec.py (proto3;
// verified against the official protobuf runtime in tests/test_schemas.py).
#include <torch/extension.h>

#include "proto_log.h"

#include <cstdint>
#include <cstring>
#include <string>
#include <vector>

namespace {

// ---- varint helpers -------------------------------------------------------

inline void put_varint(std::string& out, uint64_t v) {
  while (v >= 0x80) {
    out.push_back((char)(v | 0x80));
    v >>= 7;
  }
  out.push_back((char)v);
}

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

inline void put_tag(std::string& out, int field, int wt) {
  put_varint(out, (uint64_t)(field << 3 | wt));
}

inline void put_str(std::string& out, int field, const char* s, size_t len) {
  if (len == 0) return;
  put_tag(out, field, 2);
  put_varint(out, len);
  out.append(s, len);
}

inline void put_int(std::string& out, int field, int64_t v) {
  if (v == 0) return;
  put_tag(out, field, 0);
  put_varint(out, (uint64_t)v);  // sign-extended two's complement
}

// Overflow-safe: get_varint leaves pos <= n, so compare lengths against
// n - pos instead of computing pos + len (which a crafted ~2^64 varint
// length could wrap below n, defeating the bounds check).
bool skip_field(const uint8_t* p, size_t n, size_t& pos, int wt) {
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

// ---- decode LogSchema batch ----------------------------------------------

namespace {
using dmx_proto::LogSpan;
using dmx_proto::decode_log_core;

std::vector<std::pair<const uint8_t*, size_t>> frame_ptrs(
    const std::vector<py::bytes>& frames) {
  std::vector<std::pair<const uint8_t*, size_t>> raw(frames.size());
  for (size_t i = 0; i < frames.size(); ++i) {
    char* fptr;
    Py_ssize_t flen;
    PyBytes_AsStringAndSize(frames[i].ptr(), &fptr, &flen);
    raw[i] = {(const uint8_t*)fptr, (size_t)flen};
  }
  return raw;
}
}  // namespace

// returns (lines u8 [B, max_len], lens i32 [B], logIDs, logSources, hostnames)
py::tuple decode_log_batch(const std::vector<py::bytes>& frames,
                           int64_t max_len) {
  const int64_t B = (int64_t)frames.size();
  auto lines = torch::zeros({B, max_len}, torch::kUInt8);
  auto lens = torch::zeros({B}, torch::kInt32);
  auto raw = frame_ptrs(frames);
  std::vector<LogSpan> spans(B);
  {
    py::gil_scoped_release release;  // frames held by the caller's list
    decode_log_core(raw, max_len, lines.data_ptr<uint8_t>(),
                    lens.data_ptr<int32_t>(), spans);
  }
  py::list log_ids, sources, hostnames;
  for (int64_t i = 0; i < B; ++i) {
    log_ids.append(py::bytes(spans[i].id, spans[i].id_len));
    sources.append(py::bytes(spans[i].src, spans[i].src_len));
    hostnames.append(py::bytes(spans[i].host, spans[i].host_len));
  }
  return py::make_tuple(lines, lens, log_ids, sources, hostnames);
}

// Hot-path variant: NO per-frame Python objects. Returns
// (lines u8 [B,max_len] (pinned if pin), lens i32 [B],
//  ids_blob bytes, ids_off i32 [B+1]) — logID i is
// ids_blob[ids_off[i]:ids_off[i+1]]. Sources/hostnames are dropped
// (the fused service does not use them; decode_log_batch does).
py::tuple decode_log_batch_packed(const std::vector<py::bytes>& frames,
                                  int64_t max_len, bool pin) {
  const int64_t B = (int64_t)frames.size();
  auto opts = torch::TensorOptions().dtype(torch::kUInt8);
  if (pin) opts = opts.pinned_memory(true);
  auto lines = torch::zeros({B, max_len}, opts);
  auto lens = torch::zeros({B}, torch::TensorOptions().dtype(torch::kInt32));
  auto ids_off = torch::zeros({B + 1}, torch::TensorOptions().dtype(torch::kInt32));
  auto raw = frame_ptrs(frames);
  std::vector<LogSpan> spans(B);
  std::string blob;
  {
    py::gil_scoped_release release;
    decode_log_core(raw, max_len, lines.data_ptr<uint8_t>(),
                    lens.data_ptr<int32_t>(), spans);
    int32_t* off = ids_off.data_ptr<int32_t>();
    size_t total = 0;
    for (int64_t i = 0; i < B; ++i) total += spans[i].id_len;
    blob.reserve(total);
    for (int64_t i = 0; i < B; ++i) {
      off[i] = (int32_t)blob.size();
      blob.append(spans[i].id, spans[i].id_len);
    }
    off[B] = (int32_t)blob.size();
  }
  return py::make_tuple(lines, lens, py::bytes(blob), ids_off);
}

// ---- frame packing (send side) -------------------------------------------

// Frames -> ONE wire blob with 4-byte (intra-node) or 8-byte (SP) BE
// length prefixes; the Python dialer worker's per-frame pack+append loop
// capped the feed side of service mode (~3.3M frames/s); this is one
// GIL-released memcpy pass.
py::bytes pack_frames(const std::vector<py::bytes>& frames, bool sp) {
  std::vector<std::pair<const uint8_t*, size_t>> raw = frame_ptrs(frames);
  const size_t hdr = sp ? 8 : 4;
  std::string out;
  {
    py::gil_scoped_release release;
    size_t total = 0;
    for (auto& r : raw) total += hdr + r.second;
    out.reserve(total);
    for (auto& r : raw) {
      const uint64_t n = r.second;
      if (sp)
        for (int i = 7; i >= 0; --i) out.push_back((char)((n >> (8 * i)) & 0xFF));
      else
        for (int i = 3; i >= 0; --i) out.push_back((char)((n >> (8 * i)) & 0xFF));
      out.append((const char*)r.first, r.second);
    }
  }
  return py::bytes(out);
}

// ---- encode ParserSchema batch -------------------------------------------

// Builds serialized ParserSchema frames from the parser kernel's outputs.
// caps / fmt_caps are [B, C, 2] int32 CPU tensors of byte spans into lines.
py::list encode_parser_batch(
    torch::Tensor lines, torch::Tensor lens, torch::Tensor event_id,
    torch::Tensor caps, torch::Tensor n_caps, torch::Tensor fmt_caps,
    torch::Tensor n_fmt_caps, const std::vector<std::string>& fmt_names,
    const std::vector<std::string>& templates,
    const std::vector<py::bytes>& log_ids, const std::string& parser_type,
    const std::string& parser_id, int64_t timestamp,
    const std::string& version) {
  TORCH_CHECK(!lines.is_cuda(), "encode_parser_batch wants CPU tensors");
  const int64_t B = lines.size(0);
  const int64_t max_len = lines.size(1);
  const int64_t maxc = caps.size(1);
  const int64_t maxf = fmt_caps.size(1);
  const uint8_t* lbuf = lines.data_ptr<uint8_t>();
  const int32_t* lenp = lens.data_ptr<int32_t>();
  const int32_t* evp = event_id.data_ptr<int32_t>();
  const int32_t* cp = caps.data_ptr<int32_t>();
  const int32_t* ncp = n_caps.data_ptr<int32_t>();
  const int32_t* fp = fmt_caps.data_ptr<int32_t>();
  const int32_t* nfp = n_fmt_caps.data_ptr<int32_t>();

  py::list out;
  std::string buf;
  for (int64_t i = 0; i < B; ++i) {
    buf.clear();
    const char* line = (const char*)(lbuf + i * max_len);
    const int32_t llen = lenp[i];
    put_str(buf, 1, version.data(), version.size());
    put_str(buf, 2, parser_type.data(), parser_type.size());
    put_str(buf, 3, parser_id.data(), parser_id.size());
    const int32_t ev = evp[i];
    put_int(buf, 4, (int64_t)ev);  // sign-extended varint for negatives
    if (ev > 0 && ev <= (int32_t)templates.size())
      put_str(buf, 5, templates[ev - 1].data(), templates[ev - 1].size());
    // variables (field 6, repeated string) from capture spans
    const int32_t nc = std::min<int32_t>(ncp[i], (int32_t)maxc);
    for (int32_t j = 0; j < nc; ++j) {
      const int32_t a = cp[(i * maxc + j) * 2];
      const int32_t b = cp[(i * maxc + j) * 2 + 1];
      if (a >= 0 && b >= a && b <= llen) put_str(buf, 6, line + a, b - a);
      else put_str(buf, 6, "", 0);
    }
    // logID (8) + log (9)
    {
      char* idp;
      Py_ssize_t idl;
      PyBytes_AsStringAndSize(log_ids[i].ptr(), &idp, &idl);
      put_str(buf, 8, idp, idl);
    }
    put_str(buf, 9, line, llen);
    // logFormatVariables map (10) from format captures
    const int32_t nf = std::min<int32_t>(nfp[i], (int32_t)maxf);
    for (int32_t j = 0; j < nf && j < (int32_t)fmt_names.size(); ++j) {
      const int32_t a = fp[(i * maxf + j) * 2];
      const int32_t b = fp[(i * maxf + j) * 2 + 1];
      if (a < 0 || b < a || b > llen) continue;
      std::string entry;
      put_str(entry, 1, fmt_names[j].data(), fmt_names[j].size());
      put_str(entry, 2, line + a, b - a);
      put_tag(buf, 10, 2);
      put_varint(buf, entry.size());
      buf += entry;
    }
    put_int(buf, 11, timestamp);
    put_int(buf, 12, timestamp);
    out.append(py::bytes(buf));
  }
  return out;
}

// ---- CPU template matcher (same algorithm as template_match.hip) ----------

// Greedy-anchored wildcard match over [start, end) of `line`; returns
// capture count or -1. Mirrors match_segments in template_match.hip and
// _span_match in ops/__init__.py exactly.
int match_segments_cpu(const unsigned char* line, int start, int end,
                       const unsigned char* seg_bytes, const int32_t* seg_off,
                       int s_begin, int s_end, int32_t* caps, int max_caps,
                       bool lower) {
  int pos = start;
  int ncap = 0;
  const int nseg = s_end - s_begin;
  for (int i = 0; i < nseg; ++i) {
    const int so = seg_off[s_begin + i];
    const int sl = seg_off[s_begin + i + 1] - so;
    if (sl == 0) {
      if (i == nseg - 1) {
        if (ncap < max_caps) {
          caps[ncap * 2] = pos;
          caps[ncap * 2 + 1] = end;
        }
        return ncap + 1;
      }
      continue;
    }
    // find seg in line[pos, end)
    int idx = -1;
    for (int cand = pos; cand + sl <= end; ++cand) {
      bool ok = true;
      for (int b = 0; b < sl; ++b) {
        unsigned char c = line[cand + b];
        if (lower && c >= 'A' && c <= 'Z') c += 32;
        if (c != seg_bytes[so + b]) { ok = false; break; }
      }
      if (ok) { idx = cand; break; }
    }
    if (idx < 0) return -1;
    if (i == 0 && idx != start) return -1;
    if (i > 0) {
      if (ncap < max_caps) {
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

// CPU twin of the dmx_template_match kernel, identical outputs (the CPU
// service path was Python-matcher-bound at ~9.5k lines/s).
std::vector<torch::Tensor> template_match_cpu(
    torch::Tensor lines, torch::Tensor line_len, torch::Tensor fmt_bytes,
    torch::Tensor fmt_seg_off, torch::Tensor seg_bytes, torch::Tensor seg_off,
    torch::Tensor tpl_seg_start, bool lower, int64_t max_fmt_caps,
    int64_t max_caps) {
  TORCH_CHECK(!lines.is_cuda() && lines.dtype() == torch::kUInt8 &&
              lines.is_contiguous());
  const int64_t B = lines.size(0), max_len = lines.size(1);
  const int nf_seg =
      fmt_seg_off.numel() > 0 ? (int)fmt_seg_off.numel() - 1 : 0;
  const int n_tpl = (int)tpl_seg_start.numel() - 1;
  auto opts = torch::TensorOptions().dtype(torch::kInt32);
  auto event_id = torch::empty({B}, opts);
  auto fmt_caps = torch::zeros({B, max_fmt_caps, 2}, opts);
  auto n_fmt_caps = torch::zeros({B}, opts);
  auto caps = torch::zeros({B, max_caps, 2}, opts);
  auto n_caps = torch::zeros({B}, opts);

  const unsigned char* lbuf = lines.data_ptr<uint8_t>();
  const int32_t* lenp = line_len.data_ptr<int32_t>();
  const unsigned char* fb =
      nf_seg ? fmt_bytes.data_ptr<uint8_t>() : nullptr;
  const int32_t* fo = nf_seg ? fmt_seg_off.data_ptr<int32_t>() : nullptr;
  const unsigned char* sb = seg_bytes.data_ptr<uint8_t>();
  const int32_t* so = seg_off.data_ptr<int32_t>();
  const int32_t* ts = tpl_seg_start.data_ptr<int32_t>();
  int32_t* evp = event_id.data_ptr<int32_t>();
  int32_t* fcp = fmt_caps.data_ptr<int32_t>();
  int32_t* nfp = n_fmt_caps.data_ptr<int32_t>();
  int32_t* cp = caps.data_ptr<int32_t>();
  int32_t* ncp = n_caps.data_ptr<int32_t>();

  for (int64_t i = 0; i < B; ++i) {
    const unsigned char* line = lbuf + i * max_len;
    const int len = std::min((int)lenp[i], (int)max_len);
    int content_start = 0, content_end = len;
    int nfc = 0;
    if (nf_seg > 0) {
      nfc = match_segments_cpu(line, 0, len, fb, fo, 0, nf_seg,
                               fcp + i * max_fmt_caps * 2, (int)max_fmt_caps,
                               false);
      if (nfc > 0) {
        const int last = std::min<int>(nfc, (int)max_fmt_caps) - 1;
        content_start = fcp[i * max_fmt_caps * 2 + last * 2];
        content_end = fcp[i * max_fmt_caps * 2 + last * 2 + 1];
      } else {
        nfc = 0;
      }
    }
    nfp[i] = nfc;
    int eid = -1, nc = 0;
    for (int t = 0; t < n_tpl; ++t) {
      const int r = match_segments_cpu(line, content_start, content_end, sb,
                                       so, ts[t], ts[t + 1], cp + i * max_caps * 2,
                                       (int)max_caps, lower);
      if (r >= 0) {
        eid = t + 1;
        nc = r;
        break;
      }
    }
    evp[i] = eid;
    ncp[i] = nc;
  }
  return {event_id, fmt_caps, n_fmt_caps, caps, n_caps};
}

// ---- batched ParserSchema watch-hash extraction ---------------------------

inline uint64_t fnv1a64_host(const char* p, size_t n, bool lower) {
  uint64_t h = 1469598103934665603ull;
  for (size_t i = 0; i < n; ++i) {
    unsigned char c = (unsigned char)p[i];
    if (lower && c >= 'A' && c <= 'Z') c += 32;
    h ^= (uint64_t)c;
    h *= 1099511628211ull;
  }
  return h | 1ull;
}

// For each frame: EventID + FNV-1a hashes of the watched fields, without
// building any per-message Python object (the NewValueDetector service
// path's hot loop). var_specs = [(event, pos)] index `variables` (field 6)
// by position; hdr_names name keys of `logFormatVariables` (field 10).
// Returns (hashes i64 [B, Wv+Wh], event_ids i32 [B], logIDs list).
py::tuple parser_watch_hashes(
    const std::vector<py::bytes>& frames,
    const std::vector<std::pair<int64_t, int64_t>>& var_specs,
    const std::vector<std::string>& hdr_names, bool lower) {
  const int64_t B = (int64_t)frames.size();
  const int64_t Wv = (int64_t)var_specs.size();
  const int64_t Wh = (int64_t)hdr_names.size();
  auto hashes = torch::zeros({B, Wv + Wh}, torch::kInt64);
  auto event_ids = torch::zeros({B}, torch::kInt32);
  int64_t* hp = hashes.data_ptr<int64_t>();
  int32_t* ep = event_ids.data_ptr<int32_t>();
  py::list log_ids;

  std::vector<std::pair<const char*, size_t>> variables;
  for (int64_t i = 0; i < B; ++i) {
    char* fptr;
    Py_ssize_t flen;
    PyBytes_AsStringAndSize(frames[i].ptr(), &fptr, &flen);
    const uint8_t* p = (const uint8_t*)fptr;
    const size_t n = (size_t)flen;
    size_t pos = 0;
    int32_t event_id = 0;
    std::string log_id;
    variables.clear();
    while (pos < n) {
      uint64_t key;
      if (!get_varint(p, n, pos, key)) break;
      const int field = (int)(key >> 3), wt = (int)(key & 7);
      if (field == 4 && wt == 0) {
        uint64_t v;
        if (!get_varint(p, n, pos, v)) break;
        event_id = (int32_t)(uint32_t)v;
      } else if (wt == 2) {
        uint64_t sl;
        if (!get_varint(p, n, pos, sl) || sl > (uint64_t)(n - pos)) break;
        const char* s = (const char*)(p + pos);
        if (field == 6) {
          variables.emplace_back(s, (size_t)sl);
        } else if (field == 8) {
          log_id.assign(s, sl);
        } else if (field == 10 && Wh > 0) {
          // map entry {1: key, 2: value}
          size_t epos = 0;
          const uint8_t* e = (const uint8_t*)s;
          std::pair<const char*, size_t> k{nullptr, 0}, v{nullptr, 0};
          while (epos < sl) {
            uint64_t ekey;
            if (!get_varint(e, sl, epos, ekey)) break;
            if ((ekey & 7) == 2) {
              uint64_t el;
              if (!get_varint(e, sl, epos, el) || el > (uint64_t)(sl - epos)) break;
              if ((ekey >> 3) == 1) k = {(const char*)e + epos, (size_t)el};
              else if ((ekey >> 3) == 2) v = {(const char*)e + epos, (size_t)el};
              epos += el;
            } else if (!skip_field(e, sl, epos, (int)(ekey & 7))) {
              break;
            }
          }
          if (k.first) {
            for (int64_t w = 0; w < Wh; ++w) {
              const std::string& name = hdr_names[w];
              if (k.second == name.size() &&
                  std::memcmp(k.first, name.data(), k.second) == 0 && v.first)
                hp[i * (Wv + Wh) + Wv + w] =
                    (int64_t)fnv1a64_host(v.first, v.second, lower);
            }
          }
        }
        pos += sl;
      } else {
        if (!skip_field(p, n, pos, wt)) break;
      }
    }
    ep[i] = event_id;
    for (int64_t w = 0; w < Wv; ++w) {
      const auto& sp = var_specs[w];
      if (sp.first >= 0 && (int32_t)sp.first != event_id) continue;
      const int64_t vi = sp.second;
      if (vi >= 0 && vi < (int64_t)variables.size())
        hp[i * (Wv + Wh) + w] = (int64_t)fnv1a64_host(
            variables[vi].first, variables[vi].second, lower);
    }
    log_ids.append(py::bytes(log_id));
  }
  return py::make_tuple(hashes, event_ids, log_ids);
}

}  // namespace

void register_codec(py::module_& m) {
  m.def("pack_frames", &pack_frames, py::arg("frames"), py::arg("sp"),
        "frames -> one length-prefixed wire blob (GIL released)");
  m.def("decode_log_batch_packed", &decode_log_batch_packed,
        py::arg("frames"), py::arg("max_len"), py::arg("pin") = false,
        "hot-path LogSchema batch decode: packed ids blob, optional pinned "
        "lines buffer, GIL released during the parse");
  m.def("decode_log_batch", &decode_log_batch,
        "decode N LogSchema frames -> (lines u8 [B,max_len], lens, logIDs, "
        "sources, hostnames)");
  m.def("encode_parser_batch", &encode_parser_batch,
        "encode N ParserSchema frames from parser-kernel span outputs");
  m.def("parser_watch_hashes", &parser_watch_hashes,
        "batched ParserSchema -> watched-field FNV hashes + event ids");
  m.def("template_match_cpu", &template_match_cpu,
        "CPU twin of the dmx_template_match kernel (identical outputs)");
}
