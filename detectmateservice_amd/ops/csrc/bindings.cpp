// Torch extension bindings for the detectmate-mi355x HIP kernels.
//
// All entry points validate shapes/dtypes/contiguity, run on the CURRENT
// torch HIP stream, and are gfx950-only (no CPU fallback here — the Python
// layer in detectmateservice_amd/ops/__init__.py owns the CPU path and
// fails loudly when this extension is missing on a GPU box).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <cstdint>

extern "C" {
void dmx_launch_fused_linear_bf16(const void*, const void*, const void*,
                                  void*, int, int, int, int, hipStream_t);
void dmx_launch_probe_mfma(const void*, const void*, void*, int, int,
                           hipStream_t);
void dmx_launch_probe_mfma32(const void*, const void*, void*, hipStream_t);
void dmx_launch_layernorm_bf16(const void*, const void*, const void*,
                               const void*, void*, void*, int, int, float,
                               hipStream_t);
void dmx_launch_attention_bf16(const void*, const void*, const void*, void*,
                               int, int, int, float, hipStream_t);
void dmx_launch_attention_mfma_bf16(const void*, void*, int, int, int, int,
                                    float, hipStream_t);
void dmx_launch_bert_fused_bf16(const void*, const void*, const void*,
                                const void*, const void*, void*, int, int,
                                int, float, hipStream_t);
void dmx_launch_bert_fused_probe(const void*, const void*, const void*,
                                 const void*, const void*, void*, int, int,
                                 int, float, int, hipStream_t);
void dmx_launch_bert_fused_timed(const void*, const void*, const void*,
                                 const void*, const void*, void*, void*, int,
                                 int, int, float, hipStream_t);
void dmx_launch_template_match(const void*, const void*, int, int,
                               const void*, const void*, int, const void*,
                               int, const void*, const void*, int, int,
                               void*, void*, void*, void*, void*, void*,
                               void*, int, int, hipStream_t);
void dmx_launch_watch_hashes(const void*, int, const void*, const void*,
                             const void*, int, const void*, const void*,
                             int, const void*, int, int, int, void*,
                             hipStream_t);
void dmx_launch_hashset_insert(const void*, void*, int, int, int,
                               hipStream_t);
void dmx_launch_hashset_probe(const void*, const void*, int, int, int,
                              void*, hipStream_t);
void dmx_launch_edit_distance(const void*, const void*, int, const void*,
                              const void*, int, int, void*, hipStream_t);
}

namespace {

hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

void check_bf16_2d(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.dtype() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

torch::Tensor fused_linear_bf16(torch::Tensor x, torch::Tensor wt,
                                c10::optional<torch::Tensor> bias,
                                int64_t epilogue) {
  check_bf16_2d(x, "x");
  check_bf16_2d(wt, "wt");
  const auto M = x.size(0), K = x.size(1), N = wt.size(0);
  TORCH_CHECK(wt.size(1) == K, "wt must be [N, K] (pre-transposed weight)");
  TORCH_CHECK(K % 64 == 0, "K must be a multiple of 64");
  const void* bias_ptr = nullptr;
  if (bias.has_value() && bias->defined()) {
    TORCH_CHECK(bias->dtype() == torch::kFloat32 && bias->is_contiguous(),
                "bias must be contiguous f32");
    TORCH_CHECK(bias->numel() == N, "bias must be [N]");
    bias_ptr = bias->data_ptr();
  }
  auto C = torch::empty({M, N}, x.options());
  dmx_launch_fused_linear_bf16(x.data_ptr(), wt.data_ptr(), bias_ptr,
                               C.data_ptr(), (int)M, (int)N, (int)K,
                               (int)epilogue, cur_stream());
  return C;
}

torch::Tensor probe_mfma32(torch::Tensor A, torch::Tensor B) {
  TORCH_CHECK(A.size(0) == 32 && A.size(1) == 16, "A must be 32x16");
  TORCH_CHECK(B.size(0) == 16 && B.size(1) == 32, "B must be 16x32");
  auto D = torch::zeros({32, 32}, A.options().dtype(torch::kFloat32));
  dmx_launch_probe_mfma32(A.data_ptr(), B.data_ptr(), D.data_ptr(),
                          cur_stream());
  return D;
}

torch::Tensor probe_mfma(torch::Tensor A, torch::Tensor B, int64_t a_layout,
                         int64_t b_layout) {
  check_bf16_2d(A, "A");
  check_bf16_2d(B, "B");
  TORCH_CHECK(A.size(0) == 16 && A.size(1) == 32, "A must be 16x32");
  TORCH_CHECK(B.size(0) == 32 && B.size(1) == 16, "B must be 32x16");
  auto D = torch::zeros({16, 16},
                        A.options().dtype(torch::kFloat32));
  dmx_launch_probe_mfma(A.data_ptr(), B.data_ptr(), D.data_ptr(),
                        (int)a_layout, (int)b_layout, cur_stream());
  return D;
}

std::vector<torch::Tensor> layernorm_bf16(torch::Tensor x,
                                          c10::optional<torch::Tensor> res,
                                          torch::Tensor gamma,
                                          torch::Tensor beta, double eps,
                                          bool want_xres) {
  check_bf16_2d(x, "x");
  const auto M = x.size(0), D = x.size(1);
  TORCH_CHECK(D % 64 == 0 && D <= 2048, "D must be a multiple of 64, <=2048");
  const void* res_ptr = nullptr;
  if (res.has_value() && res->defined()) {
    check_bf16_2d(*res, "res");
    res_ptr = res->data_ptr();
  }
  auto y = torch::empty_like(x);
  torch::Tensor xres;
  void* xres_ptr = nullptr;
  if (want_xres) {
    xres = torch::empty_like(x);
    xres_ptr = xres.data_ptr();
  }
  dmx_launch_layernorm_bf16(x.data_ptr(), res_ptr, gamma.data_ptr(),
                            beta.data_ptr(), y.data_ptr(), xres_ptr, (int)M,
                            (int)D, (float)eps, cur_stream());
  if (want_xres) return {y, xres};
  return {y};
}

torch::Tensor attention_bf16(torch::Tensor q, torch::Tensor k,
                             torch::Tensor v, double scale) {
  TORCH_CHECK(q.dim() == 3, "q must be [BH, S, Dh]");
  check_bf16_2d(q.flatten(0, 1), "q");
  const auto BH = q.size(0), S = q.size(1), Dh = q.size(2);
  TORCH_CHECK(S <= 128 && Dh <= 64 && Dh % 8 == 0,
              "attention kernel supports S<=128, Dh<=64 (multiple of 8)");
  TORCH_CHECK((S * Dh) % 8 == 0, "S*Dh must be a multiple of 8");
  auto o = torch::empty_like(q);
  dmx_launch_attention_bf16(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                            o.data_ptr(), (int)BH, (int)S, (int)Dh,
                            (float)scale, cur_stream());
  return o;
}

torch::Tensor attention_qkv_bf16(torch::Tensor qkv, int64_t S, int64_t H,
                                 int64_t Dh, double scale) {
  TORCH_CHECK(qkv.is_cuda() && qkv.dtype() == torch::kBFloat16 &&
                  qkv.is_contiguous(),
              "qkv must be contiguous bf16 on GPU");
  TORCH_CHECK(qkv.dim() == 3 && qkv.size(2) == 3 * H * Dh,
              "qkv must be [B, S, 3*H*Dh]");
  TORCH_CHECK(S % 32 == 0 && S <= 128, "MFMA attention needs S%32==0, S<=128");
  TORCH_CHECK(Dh == 32 || Dh == 64, "MFMA attention needs Dh in {32,64}");
  const auto B = qkv.size(0);
  auto O = torch::empty({B, S, H * Dh}, qkv.options());
  dmx_launch_attention_mfma_bf16(qkv.data_ptr(), O.data_ptr(), (int)B, (int)S,
                                 (int)H, (int)Dh, (float)scale, cur_stream());
  return O;
}

torch::Tensor bert_fused_bf16(torch::Tensor lines, torch::Tensor start,
                              torch::Tensor end, torch::Tensor wb,
                              torch::Tensor fb, int64_t n_layers,
                              double eps) {
  TORCH_CHECK(lines.is_cuda() && lines.dtype() == torch::kUInt8 &&
                  lines.is_contiguous(),
              "lines must be contiguous u8 on GPU");
  TORCH_CHECK(wb.dtype() == torch::kBFloat16 && wb.is_contiguous());
  TORCH_CHECK(fb.dtype() == torch::kFloat32 && fb.is_contiguous());
  TORCH_CHECK(start.dtype() == torch::kInt32 && end.dtype() == torch::kInt32);
  const auto B = lines.size(0), max_len = lines.size(1);
  auto scores = torch::empty(
      {B}, torch::TensorOptions().dtype(torch::kFloat32).device(lines.device()));
  dmx_launch_bert_fused_bf16(lines.data_ptr(), start.data_ptr(),
                             end.data_ptr(), wb.data_ptr(), fb.data_ptr(),
                             scores.data_ptr(), (int)B, (int)max_len,
                             (int)n_layers, (float)eps, cur_stream());
  return scores;
}

std::vector<torch::Tensor> template_match(
    torch::Tensor lines, torch::Tensor line_len, torch::Tensor fmt_bytes,
    torch::Tensor fmt_seg_off, torch::Tensor seg_bytes, torch::Tensor seg_off,
    torch::Tensor tpl_seg_start, bool lower, int64_t max_fmt_caps,
    int64_t max_caps) {
  TORCH_CHECK(lines.is_cuda() && lines.dtype() == torch::kUInt8 &&
                  lines.is_contiguous() && lines.dim() == 2,
              "lines must be contiguous u8 [B, max_len] on GPU");
  const auto B = lines.size(0), max_len = lines.size(1);
  TORCH_CHECK(max_len <= 512, "max_len must be <= 512 (TM_MAX_LINE)");
  const int nf_seg = fmt_seg_off.numel() > 0 ? (int)fmt_seg_off.numel() - 1 : 0;
  const int n_tpl = (int)tpl_seg_start.numel() - 1;
  auto opts = torch::TensorOptions().dtype(torch::kInt32).device(lines.device());
  auto event_id = torch::empty({B}, opts);
  auto fmt_caps = torch::zeros({B, max_fmt_caps, 2}, opts);
  auto n_fmt_caps = torch::zeros({B}, opts);
  auto caps = torch::zeros({B, max_caps, 2}, opts);
  auto n_caps = torch::zeros({B}, opts);
  auto span_start = torch::empty({B}, opts);
  auto span_end = torch::empty({B}, opts);
  dmx_launch_template_match(
      lines.data_ptr(), line_len.data_ptr(), (int)B, (int)max_len,
      fmt_bytes.numel() ? fmt_bytes.data_ptr() : nullptr,
      fmt_seg_off.numel() ? fmt_seg_off.data_ptr() : nullptr, nf_seg,
      seg_bytes.data_ptr(), (int)seg_bytes.numel(), seg_off.data_ptr(),
      tpl_seg_start.data_ptr(), n_tpl, lower ? 1 : 0, event_id.data_ptr(),
      fmt_caps.data_ptr(), n_fmt_caps.data_ptr(), caps.data_ptr(),
      n_caps.data_ptr(), span_start.data_ptr(), span_end.data_ptr(),
      (int)max_fmt_caps, (int)max_caps, cur_stream());
  return {event_id, fmt_caps, n_fmt_caps, caps, n_caps, span_start, span_end};
}

torch::Tensor bert_fused_probe(torch::Tensor lines, torch::Tensor start,
                               torch::Tensor end, torch::Tensor wb,
                               torch::Tensor fb, int64_t n_layers, double eps,
                               int64_t phase_mask) {
  const auto B = lines.size(0), max_len = lines.size(1);
  auto scores = torch::empty(
      {B}, torch::TensorOptions().dtype(torch::kFloat32).device(lines.device()));
  dmx_launch_bert_fused_probe(lines.data_ptr(), start.data_ptr(),
                              end.data_ptr(), wb.data_ptr(), fb.data_ptr(),
                              scores.data_ptr(), (int)B, (int)max_len,
                              (int)n_layers, (float)eps, (int)phase_mask,
                              cur_stream());
  return scores;
}

std::vector<torch::Tensor> bert_fused_timed(torch::Tensor lines,
                                            torch::Tensor start,
                                            torch::Tensor end,
                                            torch::Tensor wb,
                                            torch::Tensor fb,
                                            int64_t n_layers, double eps) {
  const auto B = lines.size(0), max_len = lines.size(1);
  auto scores = torch::empty(
      {B}, torch::TensorOptions().dtype(torch::kFloat32).device(lines.device()));
  auto stamps = torch::zeros(
      {B, 8, 22},
      torch::TensorOptions().dtype(torch::kInt64).device(lines.device()));
  dmx_launch_bert_fused_timed(lines.data_ptr(), start.data_ptr(),
                              end.data_ptr(), wb.data_ptr(), fb.data_ptr(),
                              scores.data_ptr(), stamps.data_ptr(), (int)B,
                              (int)max_len, (int)n_layers, (float)eps,
                              cur_stream());
  return {scores, stamps};
}

torch::Tensor edit_distance(torch::Tensor A, torch::Tensor a_len,
                            torch::Tensor B, torch::Tensor b_len) {
  TORCH_CHECK(A.is_cuda() && A.dtype() == torch::kUInt8 && A.is_contiguous());
  TORCH_CHECK(B.is_cuda() && B.dtype() == torch::kUInt8 && B.is_contiguous());
  TORCH_CHECK(A.size(1) == B.size(1), "A and B must share max_len");
  TORCH_CHECK(A.size(1) <= 256, "edit_distance supports max_len<=256");
  const auto Na = A.size(0), Nb = B.size(0), max_len = A.size(1);
  auto dist = torch::empty(
      {Na, Nb}, torch::TensorOptions().dtype(torch::kInt32).device(A.device()));
  dmx_launch_edit_distance(A.data_ptr(), a_len.data_ptr(), (int)Na,
                           B.data_ptr(), b_len.data_ptr(), (int)Nb,
                           (int)max_len, dist.data_ptr(), cur_stream());
  return dist;
}

torch::Tensor watch_hashes(torch::Tensor lines, torch::Tensor event_id,
                           torch::Tensor caps, torch::Tensor n_caps,
                           torch::Tensor fmt_caps, torch::Tensor n_fmt_caps,
                           torch::Tensor specs, bool lower) {
  const auto B = lines.size(0), max_len = lines.size(1);
  const auto W = specs.size(0);
  TORCH_CHECK(specs.dtype() == torch::kInt32 && specs.size(1) == 4,
              "specs must be int32 [W, 4] (kind, event, pos, pad)");
  auto hashes = torch::zeros(
      {B, W}, torch::TensorOptions().dtype(torch::kInt64).device(lines.device()));
  dmx_launch_watch_hashes(
      lines.data_ptr(), (int)max_len, event_id.data_ptr(), caps.data_ptr(),
      n_caps.data_ptr(), (int)caps.size(1), fmt_caps.data_ptr(),
      n_fmt_caps.data_ptr(), (int)fmt_caps.size(1), specs.data_ptr(), (int)W,
      (int)B, lower ? 1 : 0, hashes.data_ptr(), cur_stream());
  return hashes;
}

void hashset_insert(torch::Tensor hashes, torch::Tensor tables) {
  const auto B = hashes.size(0), W = hashes.size(1);
  TORCH_CHECK(tables.size(0) == W, "tables must be [W, capacity]");
  const auto cap = tables.size(1);
  TORCH_CHECK((cap & (cap - 1)) == 0, "capacity must be a power of two");
  dmx_launch_hashset_insert(hashes.data_ptr(), tables.data_ptr(), (int)B,
                            (int)W, (int)cap, cur_stream());
}

torch::Tensor hashset_probe(torch::Tensor hashes, torch::Tensor tables) {
  const auto B = hashes.size(0), W = hashes.size(1);
  const auto cap = tables.size(1);
  auto unseen = torch::zeros(
      {B, W}, torch::TensorOptions().dtype(torch::kInt32).device(hashes.device()));
  dmx_launch_hashset_probe(hashes.data_ptr(), tables.data_ptr(), (int)B,
                           (int)W, (int)cap, unseen.data_ptr(), cur_stream());
  return unseen;
}

}  // namespace

void register_codec(py::module_& m);   // codec.cpp
void register_frame_reader(py::module_& m);  // frame_reader.cpp
void register_shm_ring(py::module_& m);      // shm_ring.cpp

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  register_codec(m);
  register_frame_reader(m);
  register_shm_ring(m);
  m.def("fused_linear_bf16", &fused_linear_bf16,
        "C = act(x @ wt^T + bias), bf16 MFMA (epilogue: 0 none, 1 gelu, 2 relu)");
  m.def("probe_mfma", &probe_mfma, "MFMA 16x16x32 bf16 layout probe");
  m.def("probe_mfma32", &probe_mfma32, "MFMA 32x32x16 bf16 layout probe");
  m.def("layernorm_bf16", &layernorm_bf16, "fused residual+LayerNorm bf16");
  m.def("attention_bf16", &attention_bf16, "fused short-seq MHA bf16");
  m.def("attention_qkv_bf16", &attention_qkv_bf16,
        "MFMA MHA reading fused QKV layout [B,S,3*H*Dh] -> [B,S,H*Dh]");
  m.def("bert_fused_bf16", &bert_fused_bf16,
        "whole-model BERT-tiny forward, one workgroup per line");
  m.def("bert_fused_probe", &bert_fused_probe,
        "phase-masked probe variant of the fused BERT kernel");
  m.def("bert_fused_timed", &bert_fused_timed,
        "fused kernel with per-wave phase timestamps (diagnostic)");
  m.def("template_match", &template_match, "batched wildcard template match");
  m.def("watch_hashes", &watch_hashes, "hash watched capture spans");
  m.def("hashset_insert", &hashset_insert, "insert hashes into GPU sets");
  m.def("hashset_probe", &hashset_probe, "probe hashes against GPU sets");
  m.def("edit_distance", &edit_distance,
        "batched Levenshtein distances (wavefront DP in LDS)");
}
