// Torch bindings for the npf gfx950 HIP kernels (npf._hip_C).
//
// Ops (shapes documented in npf/ops/functional.py):
//   attn_fwd(q,k,v,scale) -> (out, lse)      fp32 / bf16, per-head D<=32
//   attn_bwd(q,k,v,out,lse,dout,scale) -> (dq,dk,dv)
//   setconv_fwd(keys,queries,values,sigma) -> [B,Q,C+1]   fp32
//   setconv_bwd(...) -> (dk,dq,dv,dsigma)
//   gauss_ll_fwd(loc,scale,y) -> [rows]       fp32 (rows = Z*B)
//   gauss_ll_bwd(loc,scale,y,dout) -> (dloc,dscale)

#include <hip/hip_runtime.h>
#include <torch/extension.h>

#include <ATen/hip/HIPContext.h>

extern "C" {
void npf_attn_fwd_launch_f32(const void*, const void*, const void*, void*,
                             float*, int, int, int, int, int, float,
                             hipStream_t);
void npf_attn_fwd_launch_bf16(const void*, const void*, const void*, void*,
                              float*, int, int, int, int, int, float,
                              hipStream_t);
void npf_attn_bwd_launch_f32(const void*, const void*, const void*,
                             const void*, const float*, const void*, void*,
                             void*, void*, int, int, int, int, int, float,
                             hipStream_t);
void npf_attn_bwd_launch_bf16(const void*, const void*, const void*,
                              const void*, const float*, const void*, void*,
                              void*, void*, int, int, int, int, int, float,
                              hipStream_t);
void npf_setconv_fwd_launch(const float*, const float*, const float*, float*,
                            int, int, int, int, float, hipStream_t);
void npf_setconv_bwd_launch(const float*, const float*, const float*,
                            const float*, float*, float*, float*, float*, int,
                            int, int, int, float, hipStream_t);
void npf_gauss_ll_fwd_launch(const float*, const float*, const float*, float*,
                             long long, long long, hipStream_t);
void npf_gauss_ll_bwd_launch(const float*, const float*, const float*,
                             const float*, float*, float*, long long,
                             long long, hipStream_t);
}

namespace {

void check_cuda_contig(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

bool is_bf16(const torch::Tensor& t) {
  return t.scalar_type() == torch::kBFloat16;
}

std::tuple<torch::Tensor, torch::Tensor> attn_fwd(torch::Tensor q,
                                                  torch::Tensor k,
                                                  torch::Tensor v,
                                                  double scale) {
  check_cuda_contig(q, "q");
  check_cuda_contig(k, "k");
  check_cuda_contig(v, "v");
  const int N = q.size(0), Q = q.size(1), D = q.size(2);
  const int K = k.size(1), Dv = v.size(2);
  TORCH_CHECK(D <= 32 && Dv <= 32, "fused attn supports head dims <= 32");
  TORCH_CHECK(q.scalar_type() == k.scalar_type() &&
              q.scalar_type() == v.scalar_type());
  auto out = torch::empty({N, Q, Dv}, q.options());
  auto lse = torch::empty({N, Q}, q.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  if (is_bf16(q)) {
    npf_attn_fwd_launch_bf16(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                             out.data_ptr(), lse.data_ptr<float>(), N, Q, K, D,
                             Dv, (float)scale, stream);
  } else {
    TORCH_CHECK(q.scalar_type() == torch::kFloat32, "fp32 or bf16 only");
    npf_attn_fwd_launch_f32(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                            out.data_ptr(), lse.data_ptr<float>(), N, Q, K, D,
                            Dv, (float)scale, stream);
  }
  return {out, lse};
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> attn_bwd(
    torch::Tensor q, torch::Tensor k, torch::Tensor v, torch::Tensor out,
    torch::Tensor lse, torch::Tensor dout, double scale) {
  check_cuda_contig(q, "q");
  check_cuda_contig(k, "k");
  check_cuda_contig(v, "v");
  check_cuda_contig(out, "out");
  check_cuda_contig(lse, "lse");
  check_cuda_contig(dout, "dout");
  const int N = q.size(0), Q = q.size(1), D = q.size(2);
  const int K = k.size(1), Dv = v.size(2);
  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  auto stream = at::hip::getCurrentHIPStream();
  if (is_bf16(q)) {
    npf_attn_bwd_launch_bf16(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                             out.data_ptr(), lse.data_ptr<float>(),
                             dout.data_ptr(), dq.data_ptr(), dk.data_ptr(),
                             dv.data_ptr(), N, Q, K, D, Dv, (float)scale,
                             stream);
  } else {
    npf_attn_bwd_launch_f32(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                            out.data_ptr(), lse.data_ptr<float>(),
                            dout.data_ptr(), dq.data_ptr(), dk.data_ptr(),
                            dv.data_ptr(), N, Q, K, D, Dv, (float)scale,
                            stream);
  }
  return {dq, dk, dv};
}

torch::Tensor setconv_fwd(torch::Tensor keys, torch::Tensor queries,
                          torch::Tensor values, double sigma) {
  check_cuda_contig(keys, "keys");
  check_cuda_contig(queries, "queries");
  check_cuda_contig(values, "values");
  TORCH_CHECK(keys.scalar_type() == torch::kFloat32, "setconv is fp32");
  const int B = keys.size(0), K = keys.size(1), Q = queries.size(1),
            C = values.size(2);
  TORCH_CHECK(keys.size(2) == 1, "setconv supports x_dim == 1");
  auto out = torch::empty({B, Q, C + 1}, values.options());
  auto stream = at::hip::getCurrentHIPStream();
  npf_setconv_fwd_launch(keys.data_ptr<float>(), queries.data_ptr<float>(),
                         values.data_ptr<float>(), out.data_ptr<float>(), B, K,
                         Q, C, (float)sigma, stream);
  return out;
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor>
setconv_bwd(torch::Tensor keys, torch::Tensor queries, torch::Tensor values,
            double sigma, torch::Tensor dout) {
  check_cuda_contig(keys, "keys");
  check_cuda_contig(queries, "queries");
  check_cuda_contig(values, "values");
  check_cuda_contig(dout, "dout");
  const int B = keys.size(0), K = keys.size(1), Q = queries.size(1),
            C = values.size(2);
  auto dk = torch::zeros_like(keys);
  auto dq = torch::zeros_like(queries);
  auto dv = torch::zeros_like(values);
  auto dsigma = torch::zeros({1}, keys.options());
  auto stream = at::hip::getCurrentHIPStream();
  npf_setconv_bwd_launch(keys.data_ptr<float>(), queries.data_ptr<float>(),
                         values.data_ptr<float>(), dout.data_ptr<float>(),
                         dk.data_ptr<float>(), dq.data_ptr<float>(),
                         dv.data_ptr<float>(), dsigma.data_ptr<float>(), B, K,
                         Q, C, (float)sigma, stream);
  return {dk, dq, dv, dsigma};
}

torch::Tensor gauss_ll_fwd(torch::Tensor loc, torch::Tensor scale,
                           torch::Tensor y) {
  check_cuda_contig(loc, "loc");
  check_cuda_contig(scale, "scale");
  check_cuda_contig(y, "y");
  TORCH_CHECK(loc.scalar_type() == torch::kFloat32, "gauss_ll is fp32");
  const long long Z = loc.size(0), B = loc.size(1);
  const long long rows = Z * B;
  const long long m = loc.numel() / rows;
  auto out = torch::empty({Z, B}, loc.options());
  auto stream = at::hip::getCurrentHIPStream();
  npf_gauss_ll_fwd_launch(loc.data_ptr<float>(), scale.data_ptr<float>(),
                          y.data_ptr<float>(), out.data_ptr<float>(), rows, m,
                          stream);
  return out;
}

std::tuple<torch::Tensor, torch::Tensor> gauss_ll_bwd(torch::Tensor loc,
                                                      torch::Tensor scale,
                                                      torch::Tensor y,
                                                      torch::Tensor dout) {
  check_cuda_contig(loc, "loc");
  check_cuda_contig(scale, "scale");
  check_cuda_contig(y, "y");
  check_cuda_contig(dout, "dout");
  const long long rows = loc.size(0) * loc.size(1);
  const long long m = loc.numel() / rows;
  auto dloc = torch::empty_like(loc);
  auto dscale = torch::empty_like(scale);
  auto stream = at::hip::getCurrentHIPStream();
  npf_gauss_ll_bwd_launch(loc.data_ptr<float>(), scale.data_ptr<float>(),
                          y.data_ptr<float>(), dout.data_ptr<float>(),
                          dloc.data_ptr<float>(), dscale.data_ptr<float>(),
                          rows, m, stream);
  return {dloc, dscale};
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("attn_fwd", &attn_fwd, "fused cross-attention forward (out, lse)");
  m.def("attn_bwd", &attn_bwd, "fused cross-attention backward (dq, dk, dv)");
  m.def("setconv_fwd", &setconv_fwd, "fused Gaussian SetConv forward");
  m.def("setconv_bwd", &setconv_bwd, "fused Gaussian SetConv backward");
  m.def("gauss_ll_fwd", &gauss_ll_fwd, "fused Gaussian log-lik forward");
  m.def("gauss_ll_bwd", &gauss_ll_bwd, "fused Gaussian log-lik backward");
  m.attr("_arch") = "gfx950";
}
