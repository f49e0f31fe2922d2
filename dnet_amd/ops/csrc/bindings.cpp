// Python bindings for the dnet_amd gfx950 kernel extension.
#include <torch/extension.h>
namespace py = pybind11;

namespace dnet {
void rmsnorm_f32(torch::Tensor xf, torch::Tensor residual, torch::Tensor w,
                 torch::Tensor y, double eps);
void resid_add_f32(torch::Tensor h, torch::Tensor xf);
void rmsnorm(torch::Tensor x, c10::optional<torch::Tensor> residual,
             torch::Tensor w, torch::Tensor y, double eps);
void gemv_bf16(torch::Tensor x, torch::Tensor w, torch::Tensor out,
               c10::optional<torch::Tensor> bias);
void gemv_int8(torch::Tensor x, torch::Tensor w, torch::Tensor scales,
               torch::Tensor out, int64_t group,
               c10::optional<torch::Tensor> bias, bool packed);
void attn_decode(torch::Tensor q, torch::Tensor kcache, torch::Tensor vcache,
                 torch::Tensor pos, torch::Tensor out, double scale,
                 int64_t window, c10::optional<torch::Tensor> sinks,
                 c10::optional<torch::Tensor> kscale,
                 c10::optional<torch::Tensor> vscale,
                 c10::optional<torch::Tensor> partials, int64_t splits,
                 bool combine);
void attn_prefill(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                  c10::optional<torch::Tensor> sinks, torch::Tensor out,
                  int64_t q_off, int64_t window, double scale);
void attn_combine(torch::Tensor partials, c10::optional<torch::Tensor> sinks,
                  torch::Tensor out, int64_t splits);
void rope_append_f32(torch::Tensor src, c10::optional<torch::Tensor> bias,
                     torch::Tensor qout, int64_t Hkv, torch::Tensor kcache,
                     torch::Tensor vcache, torch::Tensor pos,
                     torch::Tensor cos_table, torch::Tensor sin_table,
                     c10::optional<torch::Tensor> kscale,
                     c10::optional<torch::Tensor> vscale,
                     c10::optional<torch::Tensor> wpos);
void rope_append(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                 torch::Tensor kcache, torch::Tensor vcache, torch::Tensor pos,
                 torch::Tensor cos_table, torch::Tensor sin_table,
                 c10::optional<torch::Tensor> kscale,
                 c10::optional<torch::Tensor> vscale,
                 c10::optional<torch::Tensor> wpos);
void swiglu(torch::Tensor gu, torch::Tensor y);
void swiglu_f32(torch::Tensor gu_f32, torch::Tensor y, int64_t N);
void dequant_int8(torch::Tensor w, torch::Tensor scales, torch::Tensor out,
                  int64_t group, bool packed);
void dequant_int4(torch::Tensor w, torch::Tensor scales, torch::Tensor out,
                  int64_t group);
void dequant_mxfp4(torch::Tensor w, torch::Tensor scales, torch::Tensor out);
bool gemm_m16_will_defer(int64_t M, int64_t N, int64_t K, int64_t group,
                         int64_t bits, int64_t scratch_elems);
bool gemm_m16(torch::Tensor x, torch::Tensor w,
              c10::optional<torch::Tensor> scales,
              c10::optional<torch::Tensor> bias, torch::Tensor out,
              c10::optional<torch::Tensor> scratch, int64_t group, bool packed,
              int64_t bits, bool defer_combine);
void moe_gateup(torch::Tensor x, torch::Tensor w,
                c10::optional<torch::Tensor> scales,
                c10::optional<torch::Tensor> bias, torch::Tensor we,
                torch::Tensor act, int64_t group, bool packed, int64_t glu,
                double alpha, double limit);
void moe_down(torch::Tensor act, torch::Tensor w,
              c10::optional<torch::Tensor> scales,
              c10::optional<torch::Tensor> bias, torch::Tensor we,
              torch::Tensor out, int64_t group, bool packed);
void col_norm2(torch::Tensor x, torch::Tensor norms);
void gather_cols(torch::Tensor x, torch::Tensor idx, torch::Tensor out);
void scatter_cols(torch::Tensor in, torch::Tensor idx, torch::Tensor out);
}  // namespace dnet

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "dnet_amd gfx950 (MI355X/CDNA4) kernels";
  m.def("rmsnorm", &dnet::rmsnorm, "fused RMSNorm (+residual)");
  m.def("gemv_bf16", &dnet::gemv_bf16, "bf16 decode GEMV");
  m.def("gemv_int8", &dnet::gemv_int8, "grouped-int8 W8A16 decode GEMV");
  m.def("attn_decode", &dnet::attn_decode, "GQA decode attention vs KV cache",
        py::arg("q"), py::arg("kcache"), py::arg("vcache"), py::arg("pos"),
        py::arg("out"), py::arg("scale"), py::arg("window"), py::arg("sinks"),
        py::arg("kscale"), py::arg("vscale"), py::arg("partials"),
        py::arg("splits"), py::arg("combine") = true);
  m.def("dequant_mxfp4", &dnet::dequant_mxfp4,
        "MXFP4 (e2m1 + e8m0 blocks) -> bf16");
  m.def("attn_prefill", &dnet::attn_prefill,
        "MFMA flash prefill attention (causal + window + sinks)",
        py::arg("q"), py::arg("k"), py::arg("v"), py::arg("sinks"),
        py::arg("out"), py::arg("q_off"), py::arg("window"),
        py::arg("scale"));
  m.def("attn_combine", &dnet::attn_combine,
        "merge split/rank flash-decode partials");
  m.def("rope_append", &dnet::rope_append, "fused RoPE + KV append (decode)",
        py::arg("q"), py::arg("k"), py::arg("v"), py::arg("kcache"),
        py::arg("vcache"), py::arg("pos"), py::arg("cos_table"),
        py::arg("sin_table"), py::arg("kscale"), py::arg("vscale"),
        py::arg("wpos") = c10::nullopt);
  m.def("swiglu", &dnet::swiglu, "fused SwiGLU");
  m.def("gemm_m16_will_defer", &dnet::gemm_m16_will_defer,
        "would gemm_m16(defer_combine=true) defer for this shape");
  m.def("rope_append_f32", &dnet::rope_append_f32,
        "RoPE+append reading+re-zeroing the qkv split-k f32 scratch");
  m.def("resid_add_f32", &dnet::resid_add_f32,
        "h += split-k f32 scratch (re-zeroed)");
  m.def("rmsnorm_f32", &dnet::rmsnorm_f32,
        "RMSNorm+residual reading+re-zeroing the split-k f32 scratch");
  m.def("swiglu_f32", &dnet::swiglu_f32,
        "SwiGLU reading+re-zeroing the split-k f32 scratch");
  m.def("dequant_int8", &dnet::dequant_int8, "grouped-int8 -> bf16 dequant");
  m.def("dequant_int4", &dnet::dequant_int4, "packed-int4 -> bf16 dequant");
  m.def("gemm_m16", &dnet::gemm_m16, "MFMA decode GEMM (M<=16, bf16 or int8)",
        py::arg("x"), py::arg("w"), py::arg("scales"), py::arg("bias"),
        py::arg("out"), py::arg("scratch"), py::arg("group"),
        py::arg("packed"), py::arg("bits"),
        py::arg("defer_combine") = false);
  m.def("moe_gateup", &dnet::moe_gateup,
        "grouped MoE gate+up GEMV with fused GLU and expert skip");
  m.def("moe_down", &dnet::moe_down,
        "grouped MoE down GEMV with weighted f32 accumulation");
  m.def("col_norm2", &dnet::col_norm2, "per-column L2 norms");
  m.def("gather_cols", &dnet::gather_cols, "pack kept columns");
  m.def("scatter_cols", &dnet::scatter_cols, "zero + scatter kept columns");
}
