// Python bindings for the modalities_amd HIP op library (gfx950-only).
#include <torch/extension.h>

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w, double eps);
std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor w, torch::Tensor invrms);
torch::Tensor rope_fwd(torch::Tensor x, torch::Tensor cos_t, torch::Tensor sin_t,
                       bool neg);
torch::Tensor rope_fwd_slice(torch::Tensor qkv, torch::Tensor cos_t,
                             torch::Tensor sin_t, long col_off, long Hn,
                             long D, bool neg);
void rope_bwd_slice(torch::Tensor dx, torch::Tensor cos_t, torch::Tensor sin_t,
                    torch::Tensor dqkv, long col_off);
torch::Tensor silu_mul_fwd(torch::Tensor g, torch::Tensor u);
std::vector<torch::Tensor> silu_mul_bwd(torch::Tensor dout, torch::Tensor g,
                                        torch::Tensor u);
torch::Tensor silu_mul_joint_fwd(torch::Tensor wv);
torch::Tensor silu_mul_joint_bwd(torch::Tensor dout, torch::Tensor wv);
std::vector<torch::Tensor> cross_entropy_fwd(torch::Tensor logits,
                                             torch::Tensor targets,
                                             long ignore_index);
torch::Tensor cross_entropy_bwd(torch::Tensor logits, torch::Tensor targets,
                                torch::Tensor lse, torch::Tensor scale,
                                long ignore_index);
void fused_adamw_masked(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                        torch::Tensor v, torch::Tensor wd_mask, double lr,
                        double beta1, double beta2, double eps, double wd,
                        double bc1, double bc2);
void fused_adamw_masked_devstep(torch::Tensor p, torch::Tensor g,
                                torch::Tensor m, torch::Tensor v,
                                torch::Tensor wd_mask, torch::Tensor step,
                                torch::Tensor bf16_out,
                                c10::optional<torch::Tensor> gscale, double lr,
                                double beta1, double beta2, double eps,
                                double wd);
void fused_adamw(std::vector<torch::Tensor> ps, std::vector<torch::Tensor> gs,
                 std::vector<torch::Tensor> ms, std::vector<torch::Tensor> vs,
                 double lr, double beta1, double beta2, double eps, double wd,
                 double bc1, double bc2);
torch::Tensor multi_tensor_sqsum(std::vector<torch::Tensor> tensors);
void multi_tensor_scale(std::vector<torch::Tensor> tensors, torch::Tensor scale);
std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, bool causal,
                                    long q_offset);
std::vector<torch::Tensor> attn_bwd(torch::Tensor dout, torch::Tensor q,
                                    torch::Tensor k, torch::Tensor v,
                                    torch::Tensor o, torch::Tensor lse,
                                    bool causal, long q_offset);
torch::Tensor mfma_probe_32x32x16(torch::Tensor a, torch::Tensor b);
torch::Tensor tr16_probe(long mode);
std::vector<torch::Tensor> attn_fwd_ablate(torch::Tensor q, torch::Tensor k,
                                           torch::Tensor v, long mode);
std::vector<torch::Tensor> attn_bwd_qkvjoint(torch::Tensor dout,
                                             torch::Tensor q, torch::Tensor k,
                                             torch::Tensor v, torch::Tensor o,
                                             torch::Tensor lse, bool causal,
                                             long q_offset, torch::Tensor dqkv,
                                             long dv_col_off);
std::vector<torch::Tensor> attn_bwd_dkdv_ablate(torch::Tensor dout,
                                                torch::Tensor q,
                                                torch::Tensor k,
                                                torch::Tensor v,
                                                torch::Tensor lse,
                                                torch::Tensor delta, long mode);
std::vector<torch::Tensor> attn_fwd2(torch::Tensor q, torch::Tensor k,
                                     torch::Tensor v, bool causal,
                                     long q_offset);
std::vector<torch::Tensor> attn_bwd2(torch::Tensor dout, torch::Tensor q,
                                     torch::Tensor k, torch::Tensor v,
                                     torch::Tensor o, torch::Tensor lse,
                                     bool causal, long q_offset);
std::vector<torch::Tensor> attn_bwd2_qkvjoint(torch::Tensor dout,
                                              torch::Tensor q, torch::Tensor k,
                                              torch::Tensor v, torch::Tensor o,
                                              torch::Tensor lse, bool causal,
                                              long q_offset, torch::Tensor dqkv,
                                              long dv_col_off);

// v1/v2 dispatch: v2 (8-wave, 2 waves/SIMD) is the perf path; v1 kept for
// A/B and as a fallback switch. Flip with set_attn_impl / MA_ATTN_IMPL.
static int g_attn_impl = [] {
  const char* e = getenv("MA_ATTN_IMPL");
  return e ? atoi(e) : 2;
}();

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm_fwd", &rmsnorm_fwd, "RMSNorm forward (K5)");
  m.def("rmsnorm_bwd", &rmsnorm_bwd, "RMSNorm backward (K5)");
  m.def("rope_fwd", &rope_fwd, "RoPE rotate-half (K4); neg=true => inverse");
  m.def("rope_fwd_slice", &rope_fwd_slice,
        "RoPE a head-block slice of fused QKV into a contiguous [B,T,H,D]");
  m.def("rope_bwd_slice", &rope_bwd_slice,
        "inverse-RoPE a [B,T,H,D] grad into a fused-QKV grad buffer slice");
  m.def("silu_mul_fwd", &silu_mul_fwd, "silu(g)*u forward (K6)");
  m.def("silu_mul_bwd", &silu_mul_bwd, "silu(g)*u backward (K6)");
  m.def("silu_mul_joint_fwd", &silu_mul_joint_fwd,
        "silu/mul over a packed [.., 2H] gate|up tensor (K6 joint)");
  m.def("silu_mul_joint_bwd", &silu_mul_joint_bwd,
        "joint-layout silu/mul backward into one dwv buffer");
  m.def("cross_entropy_fwd", &cross_entropy_fwd, "fused CE forward (K8)");
  m.def("cross_entropy_bwd", &cross_entropy_bwd, "fused CE backward (K8)");
  m.def("fused_adamw_masked", &fused_adamw_masked,
        "AdamW on a flat fp32 shard with per-element wd mask (K9)");
  m.def("fused_adamw", &fused_adamw, "multi-tensor AdamW (K9)");
  m.def("fused_adamw_masked_devstep", &fused_adamw_masked_devstep,
        "AdamW with device-resident step counter (hipGraph-safe)");
  m.def("multi_tensor_sqsum", &multi_tensor_sqsum, "sum of squares (K10)");
  m.def("multi_tensor_scale", &multi_tensor_scale, "in-place scale (K10)");
  m.def("attn_fwd",
        [](torch::Tensor q, torch::Tensor k, torch::Tensor v, bool causal,
           long q_offset) {
          const long D = q.size(3);
          return (g_attn_impl == 2 || D == 80)
              ? attn_fwd2(q, k, v, causal, q_offset)
              : attn_fwd(q, k, v, causal, q_offset);
        },
        "flash attention forward (K1)", py::arg("q"), py::arg("k"),
        py::arg("v"), py::arg("causal"), py::arg("q_offset") = 0);
  m.def("attn_bwd",
        [](torch::Tensor dout, torch::Tensor q, torch::Tensor k,
           torch::Tensor v, torch::Tensor o, torch::Tensor lse, bool causal,
           long q_offset) {
          const long D = q.size(3);
          return (g_attn_impl == 2 || D == 80)
              ? attn_bwd2(dout, q, k, v, o, lse, causal, q_offset)
              : attn_bwd(dout, q, k, v, o, lse, causal, q_offset);
        },
        "flash attention backward (K1)", py::arg("dout"), py::arg("q"),
        py::arg("k"), py::arg("v"), py::arg("o"), py::arg("lse"),
        py::arg("causal"), py::arg("q_offset") = 0);
  m.def("set_attn_impl", [](long v) { g_attn_impl = (int)v; },
        "select the K1 kernel generation (1 or 2) for A/B");
  m.def("get_attn_impl", []() { return (long)g_attn_impl; });
  m.def("attn_fwd_v1", &attn_fwd, "K1 v1 forward (direct, for A/B)");
  m.def("attn_bwd_v1", &attn_bwd, "K1 v1 backward (direct, for A/B)");
  m.def("attn_fwd_v2", &attn_fwd2, "K1 v2 forward (direct, for A/B)");
  m.def("attn_bwd_v2", &attn_bwd2, "K1 v2 backward (direct, for A/B)");
  m.def("tr16_probe", &tr16_probe, "ds_read_b64_tr_b16 semantics probe");
  m.def("mfma_probe_32x32x16", &mfma_probe_32x32x16,
        "MFMA fragment-layout probe (verification)");
  m.def("attn_fwd_ablate", &attn_fwd_ablate,
        "attention fwd cost-attribution ablation (timing only)");
  m.def("attn_bwd_qkvjoint",
        [](torch::Tensor dout, torch::Tensor q, torch::Tensor k,
           torch::Tensor v, torch::Tensor o, torch::Tensor lse, bool causal,
           long q_offset, torch::Tensor dqkv, long dv_col_off) {
          const long D = q.size(3);
          return (g_attn_impl == 2 || D == 80)
              ? attn_bwd2_qkvjoint(dout, q, k, v, o, lse, causal, q_offset,
                                   dqkv, dv_col_off)
              : attn_bwd_qkvjoint(dout, q, k, v, o, lse, causal, q_offset,
                                  dqkv, dv_col_off);
        },
        "attention backward writing dv into a fused-QKV grad buffer");
  m.def("attn_bwd_dkdv_ablate", &attn_bwd_dkdv_ablate,
        "attention bwd dkdv cost-attribution ablation (timing only)");
}
