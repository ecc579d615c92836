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

bool skip_field(const uint8_t* p, size_t n, size_t& pos, int wt) {
  uint64_t tmp;
  switch (wt) {
    case 0: return get_varint(p, n, pos, tmp);
    case 1: pos += 8; return pos <= n;
    case 2:
      if (!get_varint(p, n, pos, tmp)) return false;
      pos += tmp;
      return pos <= n;
    case 5: pos += 4; return pos <= n;
    default: return false;
  }
}

// ---- decode LogSchema batch ----------------------------------------------

// returns (lines u8 [B, max_len], lens i32 [B], logIDs, logSources, hostnames)
py::tuple decode_log_batch(const std::vector<py::bytes>& frames,
                           int64_t max_len) {
  const int64_t B = (int64_t)frames.size();
  auto lines = torch::zeros({B, max_len}, torch::kUInt8);
  auto lens = torch::zeros({B}, torch::kInt32);
  uint8_t* lbuf = lines.data_ptr<uint8_t>();
  int32_t* lenp = lens.data_ptr<int32_t>();
  py::list log_ids, sources, hostnames;

  for (int64_t i = 0; i < B; ++i) {
    char* fptr;
    Py_ssize_t flen;
    PyBytes_AsStringAndSize(frames[i].ptr(), &fptr, &flen);
    const uint8_t* p = (const uint8_t*)fptr;
    const size_t n = (size_t)flen;
    size_t pos = 0;
    std::string log_id, source, hostname;
    while (pos < n) {
      uint64_t key;
      if (!get_varint(p, n, pos, key)) break;
      const int field = (int)(key >> 3), wt = (int)(key & 7);
      if (wt == 2) {
        uint64_t sl;
        if (!get_varint(p, n, pos, sl) || pos + sl > n) break;
        const char* s = (const char*)(p + pos);
        switch (field) {
          case 2: log_id.assign(s, sl); break;
          case 3: {  // log line -> packed buffer
            const size_t copy = std::min<size_t>(sl, (size_t)max_len);
            std::memcpy(lbuf + i * max_len, s, copy);
            lenp[i] = (int32_t)copy;
            break;
          }
          case 4: source.assign(s, sl); break;
          case 5: hostname.assign(s, sl); break;
          default: break;
        }
        pos += sl;
      } else {
        if (!skip_field(p, n, pos, wt)) break;
      }
    }
    log_ids.append(py::bytes(log_id));
    sources.append(py::bytes(source));
    hostnames.append(py::bytes(hostname));
  }
  return py::make_tuple(lines, lens, log_ids, sources, hostnames);
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

}  // namespace

void register_codec(py::module_& m) {
  m.def("decode_log_batch", &decode_log_batch,
        "decode N LogSchema frames -> (lines u8 [B,max_len], lens, logIDs, "
        "sources, hostnames)");
  m.def("encode_parser_batch", &encode_parser_batch,
        "encode N ParserSchema frames from parser-kernel span outputs");
}
