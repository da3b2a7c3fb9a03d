// Torch bindings for the BurstAttention gfx950 kernels.
//
// Thin layer only: shape/stride checks, output allocation, current-stream
// plumbing.  All compute goes through the C ABI in
// include/burst_attn_hip.h (see that header for the reference call sites
// each entry point replaces).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../../include/burst_attn_hip.h"

extern "C" int bahip_mfma_probe(const void*, const void*, float*, int, void*);
extern "C" int bahip_tr16_probe(int, int*, void*);

namespace {

int dtype_code(const at::Tensor& t) {
  if (t.scalar_type() == at::kHalf) return BAHIP_F16;
  if (t.scalar_type() == at::kBFloat16) return BAHIP_BF16;
  TORCH_CHECK(false, "burst_attn: expected fp16 or bf16, got ", t.scalar_type());
  return -1;
}

void check_qkv(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be a GPU tensor");
  TORCH_CHECK(t.dim() == 4, name, " must be [B,S,N,D]");
  TORCH_CHECK(t.stride(3) == 1, name, " head_dim must be contiguous");
  TORCH_CHECK(t.stride(1) % 8 == 0, name,
              " seq stride must be a multiple of 8 elements (16B alignment)");
  TORCH_CHECK(t.stride(2) % 8 == 0, name,
              " head stride must be a multiple of 8 elements");
}

void fill_strides(const at::Tensor& t, int64_t s[3]) {
  s[0] = t.stride(0);
  s[1] = t.stride(1);
  s[2] = t.stride(2);
}

// cross-tensor consistency vs q: batch/heads/head_dim agree, k/v seqlens
// agree, dtypes are equal — a mismatch must raise here, not read out of
// bounds on device.
void check_qkv_consistent(const at::Tensor& q, const at::Tensor& k,
                          const at::Tensor& v) {
  TORCH_CHECK(k.size(0) == q.size(0) && v.size(0) == q.size(0),
              "batch mismatch");
  TORCH_CHECK(k.size(2) == q.size(2) && v.size(2) == q.size(2),
              "heads mismatch");
  TORCH_CHECK(k.size(3) == q.size(3) && v.size(3) == q.size(3),
              "head_dim mismatch");
  TORCH_CHECK(v.size(1) == k.size(1), "k/v seqlen mismatch");
  TORCH_CHECK(q.scalar_type() == k.scalar_type() &&
                  q.scalar_type() == v.scalar_type(),
              "q/k/v dtype mismatch");
}

#define BA_CALL(expr)                                                      \
  do {                                                                     \
    int rc_ = (expr);                                                      \
    TORCH_CHECK(rc_ == 0, "burst_attn HIP kernel failed (rc=", rc_, "): ", \
                bahip_last_error());                                       \
  } while (0)

std::vector<at::Tensor> attn_fwd(const at::Tensor& q, const at::Tensor& k,
                                 const at::Tensor& v, double softmax_scale,
                                 bool causal) {
  check_qkv(q, "q");
  check_qkv(k, "k");
  check_qkv(v, "v");
  const auto B = q.size(0), Sq = q.size(1), N = q.size(2), D = q.size(3);
  const auto Sk = k.size(1);
  check_qkv_consistent(q, k, v);
  auto o = at::empty({B, Sq, N, D}, q.options().dtype(at::kFloat));
  auto lse = at::empty({B, N, Sq}, q.options().dtype(at::kFloat));
  int64_t qs[3], ks[3], vs[3];
  fill_strides(q, qs);
  fill_strides(k, ks);
  fill_strides(v, vs);
  auto stream = at::hip::getCurrentHIPStream().stream();
  BA_CALL(bahip_attn_fwd(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                         o.data_ptr<float>(), lse.data_ptr<float>(), B, Sq,
                         Sk, N, D, qs, ks, vs, (float)softmax_scale,
                         causal ? 1 : 0, dtype_code(q), stream));
  return {o, lse};
}

void attn_fwd_accum(const at::Tensor& q, const at::Tensor& k,
                    const at::Tensor& v, double softmax_scale, bool causal,
                    at::Tensor& acc, at::Tensor& m, at::Tensor& l,
                    bool carry_in) {
  check_qkv(q, "q");
  check_qkv(k, "k");
  check_qkv(v, "v");
  const auto B = q.size(0), Sq = q.size(1), N = q.size(2), D = q.size(3);
  const auto Sk = k.size(1);
  check_qkv_consistent(q, k, v);
  TORCH_CHECK(acc.scalar_type() == at::kFloat && m.scalar_type() == at::kFloat
                  && l.scalar_type() == at::kFloat, "state must be fp32");
  TORCH_CHECK(acc.size(1) == Sq && m.size(2) == Sq && l.size(2) == Sq,
              "state rows must match q rows");
  TORCH_CHECK(acc.stride(3) == 1 && m.stride(2) == 1 && l.stride(2) == 1,
              "state innermost dims must be contiguous");
  int64_t qs[3], ks[3], vs[3], as[3];
  fill_strides(q, qs);
  fill_strides(k, ks);
  fill_strides(v, vs);
  fill_strides(acc, as);
  int64_t mls[2] = {m.stride(0), m.stride(1)};
  TORCH_CHECK(l.stride(0) == mls[0] && l.stride(1) == mls[1],
              "m/l stride mismatch");
  auto stream = at::hip::getCurrentHIPStream().stream();
  BA_CALL(bahip_attn_fwd_accum(
      q.data_ptr(), k.data_ptr(), v.data_ptr(), B, Sq, Sk, N, D, qs, ks, vs,
      (float)softmax_scale, causal ? 1 : 0, dtype_code(q),
      acc.data_ptr<float>(), m.data_ptr<float>(), l.data_ptr<float>(), as,
      mls, carry_in ? 1 : 0, stream));
}

std::vector<at::Tensor> attn_fwd_finalize(const at::Tensor& acc,
                                          const at::Tensor& m,
                                          const at::Tensor& l,
                                          at::ScalarType out_dtype) {
  TORCH_CHECK(acc.is_contiguous() && m.is_contiguous() && l.is_contiguous());
  const auto B = acc.size(0), S = acc.size(1), N = acc.size(2), D = acc.size(3);
  auto o = at::empty({B, S, N, D}, acc.options().dtype(out_dtype));
  auto lse = at::empty({B, N, S}, acc.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream().stream();
  BA_CALL(bahip_attn_fwd_finalize(
      acc.data_ptr<float>(), m.data_ptr<float>(), l.data_ptr<float>(),
      o.data_ptr(), lse.data_ptr<float>(), B, S, N, D,
      out_dtype == at::kHalf ? BAHIP_F16 : BAHIP_BF16, stream));
  return {o, lse};
}

at::Tensor attn_bwd_preprocess(const at::Tensor& o, const at::Tensor& dout,
                               c10::optional<at::Tensor> out) {
  check_qkv(o, "o");
  check_qkv(dout, "dout");
  const auto B = o.size(0), S = o.size(1), N = o.size(2), D = o.size(3);
  TORCH_CHECK(dout.sizes() == o.sizes(), "o/dout shape mismatch");
  at::Tensor delta;
  if (out.has_value()) {
    delta = *out;
    TORCH_CHECK(delta.scalar_type() == at::kFloat && delta.is_contiguous() &&
                    delta.sizes() == at::IntArrayRef({B, N, S}),
                "preprocess out must be contiguous fp32 [B,N,S]");
  } else {
    delta = at::empty({B, N, S}, o.options().dtype(at::kFloat));
  }
  int64_t os[3], gs[3];
  fill_strides(o, os);
  fill_strides(dout, gs);
  auto stream = at::hip::getCurrentHIPStream().stream();
  BA_CALL(bahip_attn_bwd_preprocess(o.data_ptr(), dout.data_ptr(),
                                    delta.data_ptr<float>(), B, S, N, D, os,
                                    gs, dtype_code(o), dtype_code(dout),
                                    stream));
  return delta;
}

void check_grad_out(const at::Tensor& g, const at::Tensor& like,
                    const char* name) {
  TORCH_CHECK(g.is_cuda(), name, " must be a GPU tensor");
  TORCH_CHECK(g.scalar_type() == at::kFloat, name,
              " accumulator must be fp32");
  TORCH_CHECK(g.dim() == 4, name, " must be [B,S,N,D]");
  TORCH_CHECK(g.stride(3) == 1, name, " head_dim must be contiguous");
  TORCH_CHECK(g.sizes() == like.sizes(), name, " shape mismatch");
}

// in-place accumulate: dq/dk/dv (fp32, strided views allowed) += tile
// contribution.  The ring layer's round accumulation happens HERE, not
// in python adds (reference burst_attn_interface.py:379-390).
void attn_bwd_accum(const at::Tensor& dout, const at::Tensor& q,
                    const at::Tensor& k, const at::Tensor& v,
                    const at::Tensor& delta, const at::Tensor& lse,
                    double softmax_scale, bool causal, bool deterministic,
                    at::Tensor& dq, at::Tensor& dk, at::Tensor& dv) {
  check_qkv(dout, "dout");
  check_qkv(q, "q");
  check_qkv(k, "k");
  check_qkv(v, "v");
  const auto B = q.size(0), Sq = q.size(1), N = q.size(2), D = q.size(3);
  const auto Sk = k.size(1);
  check_qkv_consistent(q, k, v);
  TORCH_CHECK(dout.size(0) == B && dout.size(1) == Sq && dout.size(2) == N &&
                  dout.size(3) == D,
              "dout/q shape mismatch");
  TORCH_CHECK(dout.scalar_type() == q.scalar_type(), "dout/q dtype mismatch");
  TORCH_CHECK(delta.scalar_type() == at::kFloat && lse.scalar_type() == at::kFloat,
              "delta/lse must be fp32");
  TORCH_CHECK(delta.dim() == 3 && lse.dim() == 3, "delta/lse must be [B,N,S]");
  TORCH_CHECK(delta.stride(2) == 1 && lse.stride(2) == 1,
              "delta/lse seq dim must be contiguous");
  TORCH_CHECK(delta.size(2) == Sq && lse.size(2) == Sq,
              "delta/lse seqlen mismatch");
  check_grad_out(dq, q, "dq");
  check_grad_out(dk, k, "dk");
  check_grad_out(dv, v, "dv");
  int64_t gs[3], qs[3], ks[3], vs[3], dqs[3], dks[3], dvs[3];
  fill_strides(dout, gs);
  fill_strides(q, qs);
  fill_strides(k, ks);
  fill_strides(v, vs);
  fill_strides(dq, dqs);
  fill_strides(dk, dks);
  fill_strides(dv, dvs);
  int64_t ds[2] = {delta.stride(0), delta.stride(1)};
  int64_t ls[2] = {lse.stride(0), lse.stride(1)};
  auto stream = at::hip::getCurrentHIPStream().stream();
  BA_CALL(bahip_attn_bwd(dout.data_ptr(), q.data_ptr(), k.data_ptr(),
                         v.data_ptr(), delta.data_ptr<float>(),
                         lse.data_ptr<float>(), dq.data_ptr<float>(),
                         dk.data_ptr<float>(), dv.data_ptr<float>(), B, Sq,
                         Sk, N, D, gs, qs, ks, vs, ds, ls, dqs, dks, dvs,
                         (float)softmax_scale, causal ? 1 : 0,
                         deterministic ? 1 : 0, dtype_code(q), stream));
}

std::vector<at::Tensor> attn_bwd(const at::Tensor& dout, const at::Tensor& q,
                                 const at::Tensor& k, const at::Tensor& v,
                                 const at::Tensor& delta,
                                 const at::Tensor& lse, double softmax_scale,
                                 bool causal, bool deterministic) {
  const auto B = q.size(0), Sq = q.size(1), N = q.size(2), D = q.size(3);
  const auto Sk = k.size(1);
  // zero-filled: the kernels accumulate, so this equals fresh outputs
  auto dq = at::zeros({B, Sq, N, D}, q.options().dtype(at::kFloat));
  auto dk = at::zeros({B, Sk, N, D}, q.options().dtype(at::kFloat));
  auto dv = at::zeros({B, Sk, N, D}, q.options().dtype(at::kFloat));
  attn_bwd_accum(dout, q, k, v, delta, lse, softmax_scale, causal,
                 deterministic, dq, dk, dv);
  return {dq, dk, dv};
}

// ---- hipModule side-load of the .s-built forward (round-3 on-ramp) ---
// attn_fwd_asm_load(path, symbol): load the hsaco once per process and
// bind the accum-form kernel; attn_fwd_accum_asm(...): same contract as
// attn_fwd_accum, dispatched through the module (kernargs packed to the
// kernel's exact C++ parameter layout; the HIP runtime appends the
// hidden arguments itself — verified by tools/asm_probe).
namespace {
hipModule_t g_asm_mod = nullptr;
std::string g_asm_path;
hipFunction_t g_asm_fn[2] = {nullptr, nullptr};  // [0]=f16, [1]=bf16

void attn_fwd_asm_load(const std::string& path, const std::string& sym_f16,
                       const std::string& sym_bf16) {
  if (g_asm_mod == nullptr || g_asm_path != path) {
    // earlier modules stay loaded (cheap); only the bound functions matter
    TORCH_CHECK(hipModuleLoad(&g_asm_mod, path.c_str()) == hipSuccess,
                "hipModuleLoad failed for ", path);
    g_asm_path = path;
  }
  TORCH_CHECK(hipModuleGetFunction(&g_asm_fn[0], g_asm_mod,
                                   sym_f16.c_str()) == hipSuccess,
              "symbol not found: ", sym_f16);
  TORCH_CHECK(hipModuleGetFunction(&g_asm_fn[1], g_asm_mod,
                                   sym_bf16.c_str()) == hipSuccess,
              "symbol not found: ", sym_bf16);
}

void attn_fwd_accum_asm(const at::Tensor& q, const at::Tensor& k,
                        const at::Tensor& v, double softmax_scale,
                        bool causal, at::Tensor& acc, at::Tensor& m,
                        at::Tensor& l, bool carry_in) {
  const int dt = dtype_code(q);
  TORCH_CHECK(g_asm_fn[dt] != nullptr, "asm forward not loaded");
  TORCH_CHECK(q.size(3) == 128, "asm forward: D=128 variants only");
  check_qkv(q, "q");
  check_qkv(k, "k");
  check_qkv(v, "v");
  check_qkv_consistent(q, k, v);
  const auto B = q.size(0), Sq = q.size(1), N = q.size(2);
  const auto Sk = k.size(1);
  // exact mirror of attn_fwd_kernel's parameter list
  struct {
    const void *q, *k, *v;
    float *o, *lse;
    int Sq, Sk, N;
    int64_t qs0, qs1, qs2, ks0, ks1, ks2, vs0, vs1, vs2;
    float scale;
    int causal;
    float *st_acc, *st_m, *st_l;
    int64_t a_sb, a_ss, a_sh, ml_sb, ml_sh;
    int carry_in;
  } args = {
      q.data_ptr(), k.data_ptr(), v.data_ptr(), nullptr, nullptr,
      (int)Sq, (int)Sk, (int)N,
      q.stride(0), q.stride(1), q.stride(2),
      k.stride(0), k.stride(1), k.stride(2),
      v.stride(0), v.stride(1), v.stride(2),
      (float)softmax_scale, causal ? 1 : 0,
      acc.data_ptr<float>(), m.data_ptr<float>(), l.data_ptr<float>(),
      acc.stride(0), acc.stride(1), acc.stride(2),
      m.stride(0), m.stride(1), carry_in ? 1 : 0,
  };
  size_t size = sizeof(args);
  void* cfg[] = {HIP_LAUNCH_PARAM_BUFFER_POINTER, &args,
                 HIP_LAUNCH_PARAM_BUFFER_SIZE, &size,
                 HIP_LAUNCH_PARAM_END};
  auto stream = at::hip::getCurrentHIPStream().stream();
  TORCH_CHECK(hipModuleLaunchKernel(
                  g_asm_fn[dt], (unsigned)((Sq + 255) / 256), (unsigned)N,
                  (unsigned)B, 512, 1, 1, 0, stream, nullptr,
                  cfg) == hipSuccess,
              "asm forward launch failed");
}
}  // namespace

at::Tensor mfma_probe(const at::Tensor& a, const at::Tensor& b) {
  TORCH_CHECK(a.is_cuda() && b.is_cuda() && a.is_contiguous() && b.is_contiguous());
  TORCH_CHECK(a.sizes() == at::IntArrayRef({32, 16}) &&
              b.sizes() == at::IntArrayRef({16, 32}));
  auto d = at::zeros({32, 32}, a.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream().stream();
  BA_CALL(bahip_mfma_probe(a.data_ptr(), b.data_ptr(), d.data_ptr<float>(),
                           dtype_code(a), stream));
  return d;
}

at::Tensor tr16_probe(int64_t mode) {
  auto out = at::zeros({256}, at::TensorOptions().dtype(at::kInt).device(at::kCUDA));
  auto stream = at::hip::getCurrentHIPStream().stream();
  BA_CALL(bahip_tr16_probe((int)mode, out.data_ptr<int>(), stream));
  return out;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("attn_fwd", &attn_fwd, "BurstAttention fwd tile (gfx950)");
  m.def("attn_fwd_accum", &attn_fwd_accum,
        "carry-in accumulator fwd tile (in-kernel LSE merge)");
  m.def("attn_fwd_finalize", &attn_fwd_finalize,
        "o = acc/l (cast), lse from state");
  m.def("attn_bwd_preprocess", &attn_bwd_preprocess, "delta = rowsum(o*do)",
        py::arg("o"), py::arg("dout"), py::arg("out") = py::none());
  m.def("attn_bwd", &attn_bwd, "BurstAttention bwd tile (gfx950)");
  m.def("attn_bwd_accum", &attn_bwd_accum,
        "bwd tile accumulating into fp32 dq/dk/dv views");
  m.def("mfma_probe", &mfma_probe, "32x32x16 MFMA layout probe");
  m.def("tr16_probe", &tr16_probe, "ds_read_tr16_b64 semantics probe");
  m.def("attn_fwd_asm_load", &attn_fwd_asm_load,
        "load the .s-built forward hsaco (round-3 on-ramp)");
  m.def("attn_fwd_accum_asm", &attn_fwd_accum_asm,
        "carry-in fwd via the hipModule-loaded .s kernel");
}
