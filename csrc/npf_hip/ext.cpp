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
void npf_attn_mfma_fwd_launch_f32(const void*, const void*, const void*,
                                  void*, float*, int, int, int, int, int,
                                  float, hipStream_t);
void npf_attn_mfma_fwd_launch_bf16(const void*, const void*, const void*,
                                   void*, float*, int, int, int, int, int,
                                   float, hipStream_t);
void npf_attn_mfma_bwd_launch_f32(const void*, const void*, const void*,
                                  const void*, const float*, const void*,
                                  float*, void*, void*, void*, int, int, int,
                                  int, int, float, hipStream_t);
void npf_attn_mfma_bwd_launch_bf16(const void*, const void*, const void*,
                                   const void*, const float*, const void*,
                                   float*, void*, void*, void*, int, int,
                                   int, int, int, float, hipStream_t);
void npf_setconv_fwd_launch(const float*, const float*, const float*, float*,
                            int, int, int, int, const float*, hipStream_t);
void npf_setconv_bwd_launch(const float*, const float*, const float*,
                            const float*, float*, float*, float*, float*, int,
                            int, int, int, const float*, hipStream_t);
void npf_gauss_ll_fwd_launch(const float*, const float*, const float*, float*,
                             long long, long long, hipStream_t);
void npf_gauss_ll_bwd_launch(const float*, const float*, const float*,
                             const float*, float*, float*, long long,
                             long long, hipStream_t);
void npf_lse_z_fwd_launch(const float*, float*, long long, long long,
                          hipStream_t);
void npf_lse_z_bwd_launch(const float*, const float*, const float*, float*,
                          long long, long long, hipStream_t);
void npf_qkv_fwd_launch(const void* const*, const float* const*,
                        const float* const*, void* const*, const long*,
                        const int*, const int*, int, int, int, hipStream_t);
void npf_qkv_bwd_launch(const void* const*, void* const*,
                        const float* const*, void* const*, float* const*,
                        const long*, const int*, const int*, int, int, int,
                        hipStream_t);
void npf_add_ln_fwd_launch(const void*, const void*, const float*,
                           const float*, void*, void*, float*, float*, long,
                           int, int, int, int, float, hipStream_t);
void npf_add_ln_bwd_launch(const void*, const void*, const float*,
                           const float*, const float*, void*, void*, float*,
                           float*, long, int, int, int, int, hipStream_t);
void npf_cb_stats_launch(const void*, int, float*, float*, float*, float*,
                         float*, float*, int, int, int, float, float,
                         hipStream_t);
void npf_cb_fwd_launch(const float*, const float*, const float*, const float*,
                       const float*, const float*, const float*, const float*,
                       float*, int, int, int, int, hipStream_t);
void npf_cb_bwd_launch(const float*, const float*, const float*, const float*,
                       const float*, const float*, const float*, float*,
                       float*, float*, float*, float*, float*, float*, float*,
                       int, int, int, int, int, hipStream_t);
void npf_cb_bwd_dx_launch(const void*, const void*, const void*,
                          const float*, const float*, const float*,
                          const float*, const float*, void*, int, int, int,
                          int, int, hipStream_t);
void npf_cb2d_fwd_launch(const void*, const void*, const float*,
                         const float*, const float*, const float*,
                         const float*, const float*, void*, int, int, int,
                         int, int, int, hipStream_t);
void npf_cb2d_bwd_dact_launch(const void*, const float*, const void*,
                              const float*, const float*, const float*,
                              const float*, void*, float*, float*, float*,
                              float*, float*, float*, int, int, int, int,
                              int, int, hipStream_t);
void npf_gde_fwd_launch(const float*, const float*, const float*, float*, int,
                        int, int, int, int, hipStream_t);
void npf_gde_bwd_launch(const float*, const float*, const float*, const float*,
                        const float*, float*, float*, int, int, int, int, int,
                        hipStream_t);
void npf_gauss_kl_fwd_launch(const float*, const float*, const float*,
                             const float*, float*, long long, long long,
                             hipStream_t);
void npf_gauss_kl_bwd_launch(const float*, const float*, const float*,
                             const float*, const float*, float*, float*,
                             float*, float*, long long, long long,
                             hipStream_t);
void npf_mlp_fwd_launch(const void*, const float* const*, const float* const*,
                        void* const*, void*, const int*, int, long,
                        hipStream_t);
void npf_mlp_bwd_launch(const void*, const float* const*, const void* const*,
                        void* const*, void*, float* const*, const int*, int,
                        long, hipStream_t);
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
  // large-K bf16 -> MFMA flash kernel (attn_mfma.hip).  fp32 inputs keep
  // the VALU kernel (exact fp32 math — MFMA would silently round operands
  // to bf16); tiny K also stays VALU (MFMA tiles would be mostly padding).
  const bool mfma = K >= 96 && is_bf16(q);
  if (is_bf16(q)) {
    (mfma ? npf_attn_mfma_fwd_launch_bf16 : npf_attn_fwd_launch_bf16)(
        q.data_ptr(), k.data_ptr(), v.data_ptr(), out.data_ptr(),
        lse.data_ptr<float>(), N, Q, K, D, Dv, (float)scale, stream);
  } else {
    TORCH_CHECK(q.scalar_type() == torch::kFloat32, "fp32 or bf16 only");
    (mfma ? npf_attn_mfma_fwd_launch_f32 : npf_attn_fwd_launch_f32)(
        q.data_ptr(), k.data_ptr(), v.data_ptr(), out.data_ptr(),
        lse.data_ptr<float>(), N, Q, K, D, Dv, (float)scale, stream);
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
  const bool mfma = K >= 96 && is_bf16(q);
  if (mfma) {
    auto delta = torch::empty({N, Q}, lse.options());
    if (is_bf16(q))
      npf_attn_mfma_bwd_launch_bf16(
          q.data_ptr(), k.data_ptr(), v.data_ptr(), out.data_ptr(),
          lse.data_ptr<float>(), dout.data_ptr(), delta.data_ptr<float>(),
          dq.data_ptr(), dk.data_ptr(), dv.data_ptr(), N, Q, K, D, Dv,
          (float)scale, stream);
    else
      npf_attn_mfma_bwd_launch_f32(
          q.data_ptr(), k.data_ptr(), v.data_ptr(), out.data_ptr(),
          lse.data_ptr<float>(), dout.data_ptr(), delta.data_ptr<float>(),
          dq.data_ptr(), dk.data_ptr(), dv.data_ptr(), N, Q, K, D, Dv,
          (float)scale, stream);
  } else if (is_bf16(q)) {
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
                          torch::Tensor values, torch::Tensor sigma) {
  check_cuda_contig(sigma, "sigma");
  TORCH_CHECK(sigma.scalar_type() == torch::kFloat32, "sigma must be fp32");
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
                         Q, C, sigma.data_ptr<float>(), stream);
  return out;
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor>
setconv_bwd(torch::Tensor keys, torch::Tensor queries, torch::Tensor values,
            torch::Tensor sigma, torch::Tensor dout) {
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
                         Q, C, sigma.data_ptr<float>(), stream);
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

// NPML epilogue: logsumexp over the z dim of a [Z, B] matrix, minus log Z
torch::Tensor lse_z_fwd(torch::Tensor w) {
  check_cuda_contig(w, "w");
  TORCH_CHECK(w.dim() == 2 && w.scalar_type() == torch::kFloat32,
              "lse_z expects fp32 [Z, B]");
  const long long Z = w.size(0), B = w.size(1);
  auto out = torch::empty({B}, w.options());
  auto stream = at::hip::getCurrentHIPStream();
  npf_lse_z_fwd_launch(w.data_ptr<float>(), out.data_ptr<float>(), Z, B,
                       stream);
  return out;
}

torch::Tensor lse_z_bwd(torch::Tensor w, torch::Tensor out,
                        torch::Tensor dout) {
  check_cuda_contig(w, "w");
  check_cuda_contig(out, "out");
  check_cuda_contig(dout, "dout");
  const long long Z = w.size(0), B = w.size(1);
  auto dw = torch::empty_like(w);
  auto stream = at::hip::getCurrentHIPStream();
  npf_lse_z_bwd_launch(w.data_ptr<float>(), out.data_ptr<float>(),
                       dout.data_ptr<float>(), dw.data_ptr<float>(), Z, B,
                       stream);
  return dw;
}

// Fused K/Q/V projections with head-split store: for each problem i,
// out_i = headsplit(x_i @ w_i^T + b_i) with layout [H*B_i, N_i, D/H]
std::vector<torch::Tensor> qkv_fwd(std::vector<torch::Tensor> xs,
                                   std::vector<torch::Tensor> ws,
                                   std::vector<torch::Tensor> bs,
                                   std::vector<int64_t> Bs,
                                   std::vector<int64_t> Ns, int64_t H) {
  const int n = (int)xs.size();
  TORCH_CHECK(n >= 1 && n <= 3, "qkv: 1..3 problems");
  const int D = (int)xs[0].size(-1);
  TORCH_CHECK(D <= 128 && D % (int)(16 * H) == 0,
              "qkv: D<=128 and head size multiple of 16");
  const void* xp[3] = {};
  const float* wp[3] = {};
  const float* bp[3] = {};
  void* op[3] = {};
  long Rs[3] = {};
  int Bi[3] = {}, Ni[3] = {};
  std::vector<torch::Tensor> outs;
  for (int i = 0; i < n; ++i) {
    check_cuda_contig(xs[i], "qkv x");
    check_cuda_contig(ws[i], "qkv w");
    TORCH_CHECK(xs[i].scalar_type() == torch::kBFloat16, "qkv x bf16");
    TORCH_CHECK(ws[i].scalar_type() == torch::kFloat32, "qkv w fp32");
    TORCH_CHECK(ws[i].size(0) == D && ws[i].size(1) == D, "qkv square W");
    TORCH_CHECK((long)Bs[i] * Ns[i] == xs[i].numel() / D, "qkv B*N");
    auto out = torch::empty({H * Bs[i], Ns[i], D / H},
                            xs[i].options());
    outs.push_back(out);
    xp[i] = xs[i].data_ptr();
    wp[i] = ws[i].data_ptr<float>();
    bp[i] = (bs[i].defined() && bs[i].numel() > 0) ? bs[i].data_ptr<float>()
                                                   : nullptr;
    op[i] = out.data_ptr();
    Rs[i] = (long)Bs[i] * Ns[i];
    Bi[i] = (int)Bs[i];
    Ni[i] = (int)Ns[i];
  }
  auto stream = at::hip::getCurrentHIPStream();
  npf_qkv_fwd_launch(xp, wp, bp, op, Rs, Bi, Ni, n, D, (int)H, stream);
  return outs;
}

// backward: returns per-problem (dx [R,D] bf16, dz [R,D] bf16 row-major)
// plus db fp32 for problems flagged with bias; the dW GEMMs run in python
std::vector<torch::Tensor> qkv_bwd(std::vector<torch::Tensor> douts,
                                   std::vector<torch::Tensor> ws,
                                   std::vector<int64_t> Bs,
                                   std::vector<int64_t> Ns,
                                   std::vector<int64_t> has_bias, int64_t H) {
  const int n = (int)douts.size();
  const int D = (int)ws[0].size(0);
  const void* dop[3] = {};
  void* dzp[3] = {};
  const float* wp[3] = {};
  void* dxp[3] = {};
  float* dbp[3] = {};
  long Rs[3] = {};
  int Bi[3] = {}, Ni[3] = {};
  std::vector<torch::Tensor> rets;
  for (int i = 0; i < n; ++i) {
    check_cuda_contig(douts[i], "qkv dout");
    const long R = (long)Bs[i] * Ns[i];
    auto dx = torch::empty({R, (long)D}, douts[i].options());
    auto dz = torch::empty({R, (long)D}, douts[i].options());
    torch::Tensor db;
    if (has_bias[i]) {
      db = torch::zeros({(long)D}, ws[i].options());
      dbp[i] = db.data_ptr<float>();
    }
    rets.push_back(dx);
    rets.push_back(dz);
    rets.push_back(db);
    dop[i] = douts[i].data_ptr();
    dzp[i] = dz.data_ptr();
    wp[i] = ws[i].data_ptr<float>();
    dxp[i] = dx.data_ptr();
    Rs[i] = R;
    Bi[i] = (int)Bs[i];
    Ni[i] = (int)Ns[i];
  }
  auto stream = at::hip::getCurrentHIPStream();
  npf_qkv_bwd_launch(dop, dzp, wp, dxp, dbp, Rs, Bi, Ni, n, D, (int)H,
                     stream);
  return rets;
}

// y = LayerNorm(a + b); a optionally in head-split layout (H != 0)
std::vector<torch::Tensor> add_ln_fwd(torch::Tensor a, torch::Tensor b,
                                      torch::Tensor gamma, torch::Tensor beta,
                                      int64_t B, int64_t N, int64_t H,
                                      double eps) {
  check_cuda_contig(a, "aln a");
  check_cuda_contig(b, "aln b");
  TORCH_CHECK(a.scalar_type() == torch::kBFloat16 &&
                  b.scalar_type() == torch::kBFloat16,
              "add_ln bf16 inputs");
  const int D = (int)b.size(-1);
  TORCH_CHECK(D <= 128, "add_ln D<=128");
  const long R = b.numel() / D;
  auto y = torch::empty_like(b.reshape({R, (long)D}));
  auto s = torch::empty_like(y);
  auto mean = torch::empty({R}, gamma.options());
  auto rstd = torch::empty({R}, gamma.options());
  auto stream = at::hip::getCurrentHIPStream();
  npf_add_ln_fwd_launch(a.data_ptr(), b.data_ptr(),
                        gamma.data_ptr<float>(), beta.data_ptr<float>(),
                        y.data_ptr(), s.data_ptr(), mean.data_ptr<float>(),
                        rstd.data_ptr<float>(), R, D, (int)B, (int)N, (int)H,
                        (float)eps, stream);
  return {y, s, mean, rstd};
}

std::vector<torch::Tensor> add_ln_bwd(torch::Tensor s, torch::Tensor dy,
                                      torch::Tensor gamma, torch::Tensor mean,
                                      torch::Tensor rstd, int64_t B,
                                      int64_t N, int64_t H) {
  check_cuda_contig(s, "aln s");
  check_cuda_contig(dy, "aln dy");
  const int D = (int)s.size(-1);
  const long R = s.numel() / D;
  torch::Tensor da;
  if (H != 0)
    da = torch::empty({H * B, N, (long)(D / H)}, s.options());
  else
    da = torch::empty_like(s);
  auto db = torch::empty_like(s);
  auto dgamma = torch::zeros_like(gamma);
  auto dbeta = torch::zeros_like(gamma);
  auto stream = at::hip::getCurrentHIPStream();
  npf_add_ln_bwd_launch(s.data_ptr(), dy.data_ptr(),
                        gamma.data_ptr<float>(), mean.data_ptr<float>(),
                        rstd.data_ptr<float>(), da.data_ptr(), db.data_ptr(),
                        dgamma.data_ptr<float>(), dbeta.data_ptr<float>(), R,
                        D, (int)B, (int)N, (int)H, stream);
  return {da, db, dgamma, dbeta};
}

// fused conv block: stats (training BN) -> (mean, rstd, save_var)
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> convblock_stats(
    torch::Tensor x, torch::Tensor running_mean, torch::Tensor running_var,
    double eps, double momentum) {
  check_cuda_contig(x, "x");
  const int N = x.size(0), C = x.size(1), L = x.size(2);
  auto fopts = x.options().dtype(torch::kFloat32);
  auto mean = torch::empty({C}, fopts);
  auto rstd = torch::empty({C}, fopts);
  auto save_var = torch::empty({C}, fopts);
  auto sums = torch::zeros({2 * C}, fopts);  // (sum ; sumsq) workspace
  auto stream = at::hip::getCurrentHIPStream();
  float* rm = running_mean.defined() ? running_mean.data_ptr<float>() : nullptr;
  float* rv = running_var.defined() ? running_var.data_ptr<float>() : nullptr;
  npf_cb_stats_launch(x.data_ptr(), is_bf16(x) ? 1 : 0,
                      sums.data_ptr<float>(), mean.data_ptr<float>(),
                      rstd.data_ptr<float>(), save_var.data_ptr<float>(), rm,
                      rv, N, C, L, (float)eps, (float)momentum, stream);
  return {mean, rstd, save_var};
}

torch::Tensor convblock_fwd(torch::Tensor x, torch::Tensor res,
                            torch::Tensor w, torch::Tensor b,
                            torch::Tensor gamma, torch::Tensor beta,
                            torch::Tensor mean, torch::Tensor rstd) {
  check_cuda_contig(x, "x");
  check_cuda_contig(w, "w");
  TORCH_CHECK(x.scalar_type() == torch::kFloat32, "convblock is fp32");
  const int N = x.size(0), C = x.size(1), L = x.size(2);
  const int K = w.size(-1);
  TORCH_CHECK(K <= 31 && K % 2 == 1, "kernel size must be odd and <= 31");
  auto y = torch::empty_like(x);
  auto stream = at::hip::getCurrentHIPStream();
  npf_cb_fwd_launch(
      x.data_ptr<float>(), res.defined() ? res.data_ptr<float>() : nullptr,
      w.data_ptr<float>(),
      b.defined() ? b.data_ptr<float>() : nullptr,
      gamma.defined() ? gamma.data_ptr<float>() : nullptr,
      gamma.defined() ? beta.data_ptr<float>() : nullptr,
      gamma.defined() ? mean.data_ptr<float>() : nullptr,
      gamma.defined() ? rstd.data_ptr<float>() : nullptr, y.data_ptr<float>(),
      N, C, L, K, stream);
  return y;
}

std::vector<torch::Tensor> convblock_bwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor dy, torch::Tensor gamma,
                                         torch::Tensor beta, torch::Tensor mean,
                                         torch::Tensor rstd, bool has_bias,
                                         bool training) {
  check_cuda_contig(x, "x");
  check_cuda_contig(w, "w");
  check_cuda_contig(dy, "dy");
  const int N = x.size(0), C = x.size(1), L = x.size(2);
  const int K = w.size(-1);
  const bool has_bn = gamma.defined();
  auto dact = torch::empty_like(x);
  auto dx = torch::empty_like(x);
  auto dw = torch::zeros_like(w);
  auto db = has_bias ? torch::zeros({C}, x.options()) : torch::Tensor();
  auto opts = x.options();
  auto sum_dxhat = has_bn ? torch::zeros({C}, opts) : torch::Tensor();
  auto sum_dxhat_xhat = has_bn ? torch::zeros({C}, opts) : torch::Tensor();
  auto dgamma = has_bn ? torch::zeros({C}, opts) : torch::Tensor();
  auto dbeta = has_bn ? torch::zeros({C}, opts) : torch::Tensor();
  auto stream = at::hip::getCurrentHIPStream();
  npf_cb_bwd_launch(
      x.data_ptr<float>(), w.data_ptr<float>(), dy.data_ptr<float>(),
      has_bn ? gamma.data_ptr<float>() : nullptr,
      has_bn ? beta.data_ptr<float>() : nullptr,
      has_bn ? mean.data_ptr<float>() : nullptr,
      has_bn ? rstd.data_ptr<float>() : nullptr, dact.data_ptr<float>(),
      dw.data_ptr<float>(), has_bias ? db.data_ptr<float>() : nullptr,
      has_bn ? sum_dxhat.data_ptr<float>() : nullptr,
      has_bn ? sum_dxhat_xhat.data_ptr<float>() : nullptr,
      has_bn ? dgamma.data_ptr<float>() : nullptr,
      has_bn ? dbeta.data_ptr<float>() : nullptr, dx.data_ptr<float>(), N, C,
      L, K, training ? 1 : 0, stream);
  return {dx, dw, db, dgamma, dbeta};
}


// 2D fused conv block (GridConv models); stats reuse convblock_stats with
// x viewed as [N, C, H*W]
torch::Tensor convblock2d_fwd(torch::Tensor x, torch::Tensor res,
                              torch::Tensor w, torch::Tensor b,
                              torch::Tensor gamma, torch::Tensor beta,
                              torch::Tensor mean, torch::Tensor rstd) {
  check_cuda_contig(x, "x");
  check_cuda_contig(w, "w");
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int K = w.size(-1);
  TORCH_CHECK(K <= 13 && K % 2 == 1, "2D kernel size must be odd and <= 13");
  auto y = torch::empty_like(x);
  auto stream = at::hip::getCurrentHIPStream();
  npf_cb2d_fwd_launch(
      x.data_ptr(), res.defined() ? res.data_ptr() : nullptr,
      w.data_ptr<float>(), b.defined() ? b.data_ptr<float>() : nullptr,
      gamma.defined() ? gamma.data_ptr<float>() : nullptr,
      gamma.defined() ? beta.data_ptr<float>() : nullptr,
      gamma.defined() ? mean.data_ptr<float>() : nullptr,
      gamma.defined() ? rstd.data_ptr<float>() : nullptr, y.data_ptr(), N, C,
      H, W, K, is_bf16(x) ? 1 : 0, stream);
  return y;
}

std::vector<torch::Tensor> convblock2d_bwd(torch::Tensor x, torch::Tensor w,
                                           torch::Tensor dy,
                                           torch::Tensor gamma,
                                           torch::Tensor beta,
                                           torch::Tensor mean,
                                           torch::Tensor rstd, bool has_bias,
                                           bool training) {
  check_cuda_contig(x, "x");
  check_cuda_contig(w, "w");
  check_cuda_contig(dy, "dy");
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int K = w.size(-1);
  const bool has_bn = gamma.defined();
  const int bf = is_bf16(x) ? 1 : 0;
  auto dact = torch::empty_like(x);
  auto dx = torch::empty_like(x);
  auto dw = torch::zeros_like(w);
  auto opts = x.options().dtype(torch::kFloat32);
  auto db = has_bias ? torch::zeros({C}, opts) : torch::Tensor();
  auto sum_dxhat = has_bn ? torch::zeros({C}, opts) : torch::Tensor();
  auto sum_dxhat_xhat = has_bn ? torch::zeros({C}, opts) : torch::Tensor();
  auto dgamma = has_bn ? torch::zeros({C}, opts) : torch::Tensor();
  auto dbeta = has_bn ? torch::zeros({C}, opts) : torch::Tensor();
  auto stream = at::hip::getCurrentHIPStream();
  npf_cb2d_bwd_dact_launch(
      x.data_ptr(), w.data_ptr<float>(), dy.data_ptr(),
      has_bn ? gamma.data_ptr<float>() : nullptr,
      has_bn ? beta.data_ptr<float>() : nullptr,
      has_bn ? mean.data_ptr<float>() : nullptr,
      has_bn ? rstd.data_ptr<float>() : nullptr, dact.data_ptr(),
      dw.data_ptr<float>(), has_bias ? db.data_ptr<float>() : nullptr,
      has_bn ? sum_dxhat.data_ptr<float>() : nullptr,
      has_bn ? sum_dxhat_xhat.data_ptr<float>() : nullptr,
      has_bn ? dgamma.data_ptr<float>() : nullptr,
      has_bn ? dbeta.data_ptr<float>() : nullptr, N, C, H, W, K, bf, stream);
  npf_cb_bwd_dx_launch(
      x.data_ptr(), dact.data_ptr(), dy.data_ptr(),
      has_bn ? gamma.data_ptr<float>() : nullptr,
      has_bn ? mean.data_ptr<float>() : nullptr,
      has_bn ? rstd.data_ptr<float>() : nullptr,
      has_bn ? sum_dxhat.data_ptr<float>() : nullptr,
      has_bn ? sum_dxhat_xhat.data_ptr<float>() : nullptr, dx.data_ptr(), N,
      C, H * W, training ? 1 : 0, bf, stream);
  return {dx, dw, db, dgamma, dbeta};
}


// grid density encoder (GridConv masked abs-conv pair + divide + concat)
torch::Tensor griddensity_fwd(torch::Tensor x, torch::Tensor m,
                              torch::Tensor w) {
  check_cuda_contig(x, "x");
  check_cuda_contig(m, "m");
  check_cuda_contig(w, "w");
  TORCH_CHECK(x.scalar_type() == torch::kFloat32, "griddensity is fp32");
  const int B = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int K = w.size(-1);
  TORCH_CHECK(K <= 13 && K % 2 == 1, "kernel size must be odd and <= 13");
  auto out = torch::empty({B, 2 * C, H, W}, x.options());
  auto stream = at::hip::getCurrentHIPStream();
  npf_gde_fwd_launch(x.data_ptr<float>(), m.data_ptr<float>(),
                     w.data_ptr<float>(), out.data_ptr<float>(), B, C, H, W,
                     K, stream);
  return out;
}

std::tuple<torch::Tensor, torch::Tensor> griddensity_bwd(torch::Tensor x,
                                                         torch::Tensor m,
                                                         torch::Tensor dout,
                                                         torch::Tensor out,
                                                         torch::Tensor w) {
  check_cuda_contig(dout, "dout");
  check_cuda_contig(out, "out");
  const int B = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int K = w.size(-1);
  auto dx = torch::empty_like(x);
  auto dw = torch::empty_like(w);
  auto stream = at::hip::getCurrentHIPStream();
  npf_gde_bwd_launch(x.data_ptr<float>(), m.data_ptr<float>(),
                     dout.data_ptr<float>(), out.data_ptr<float>(),
                     w.data_ptr<float>(), dx.data_ptr<float>(),
                     dw.data_ptr<float>(), B, C, H, W, K, stream);
  return {dx, dw};
}


// fused diagonal-Gaussian KL + reduce: [B, ...] -> [B]
torch::Tensor gauss_kl_fwd(torch::Tensor mq, torch::Tensor sq,
                           torch::Tensor mp, torch::Tensor sp) {
  for (auto* t : {&mq, &sq, &mp, &sp}) check_cuda_contig(*t, "kl input");
  TORCH_CHECK(mq.scalar_type() == torch::kFloat32, "gauss_kl is fp32");
  // the kernel indexes all four tensors with mq's layout: reject
  // broadcastable (size-1) dims instead of reading out of bounds
  TORCH_CHECK(sq.sizes() == mq.sizes() && mp.sizes() == mq.sizes() &&
                  sp.sizes() == mq.sizes(),
              "gauss_kl requires equal shapes for mq/sq/mp/sp");
  const long long rows = mq.size(0);
  const long long m = mq.numel() / rows;
  auto out = torch::empty({rows}, mq.options());
  auto stream = at::hip::getCurrentHIPStream();
  npf_gauss_kl_fwd_launch(mq.data_ptr<float>(), sq.data_ptr<float>(),
                          mp.data_ptr<float>(), sp.data_ptr<float>(),
                          out.data_ptr<float>(), rows, m, stream);
  return out;
}

std::vector<torch::Tensor> gauss_kl_bwd(torch::Tensor mq, torch::Tensor sq,
                                        torch::Tensor mp, torch::Tensor sp,
                                        torch::Tensor dout) {
  const long long rows = mq.size(0);
  const long long m = mq.numel() / rows;
  auto dmq = torch::empty_like(mq);
  auto dsq = torch::empty_like(sq);
  auto dmp = torch::empty_like(mp);
  auto dsp = torch::empty_like(sp);
  auto stream = at::hip::getCurrentHIPStream();
  npf_gauss_kl_bwd_launch(mq.data_ptr<float>(), sq.data_ptr<float>(),
                          mp.data_ptr<float>(), sp.data_ptr<float>(),
                          dout.data_ptr<float>(), dmq.data_ptr<float>(),
                          dsq.data_ptr<float>(), dmp.data_ptr<float>(),
                          dsp.data_ptr<float>(), rows, m, stream);
  return {dmq, dsq, dmp, dsp};
}


// fused MFMA MLP chain: x [R, d0] bf16 -> y [R, dL] bf16 (+ saved acts)
std::vector<torch::Tensor> mlp_chain_fwd(torch::Tensor x,
                                         std::vector<torch::Tensor> ws,
                                         std::vector<torch::Tensor> bs) {
  check_cuda_contig(x, "x");
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "mlp_chain x must be bf16");
  const int L = (int)ws.size();
  TORCH_CHECK(L >= 2 && L <= 8, "mlp_chain supports 2..8 layers");
  const long R = x.numel() / x.size(-1);
  int d[9];
  d[0] = (int)x.size(-1);
  const float* wp[8];
  const float* bp[8];
  void* ap[8] = {nullptr};
  std::vector<torch::Tensor> acts;
  for (int i = 0; i < L; ++i) {
    check_cuda_contig(ws[i], "w");
    TORCH_CHECK(ws[i].scalar_type() == torch::kFloat32, "weights must be fp32");
    TORCH_CHECK(ws[i].size(1) == d[i], "layer dim mismatch");
    d[i + 1] = (int)ws[i].size(0);
    TORCH_CHECK(d[i + 1] <= 128 && d[i] <= 128, "mlp_chain dims <= 128");
    wp[i] = ws[i].data_ptr<float>();
    bp[i] = bs[i].data_ptr<float>();
    if (i < L - 1) {
      acts.push_back(torch::empty({R, (long)d[i + 1]},
                                  x.options().dtype(torch::kBFloat16)));
      ap[i] = acts.back().data_ptr();
    }
  }
  auto y = torch::empty({R, (long)d[L]}, x.options().dtype(torch::kBFloat16));
  auto stream = at::hip::getCurrentHIPStream();
  npf_mlp_fwd_launch(x.data_ptr(), wp, bp, ap, y.data_ptr(), d, L, R, stream);
  std::vector<torch::Tensor> out = {y};
  for (auto& a : acts) out.push_back(a);
  return out;
}

// backward: returns {dx_or_empty, dz_0..dz_{L-1}, db_0..db_{L-1}}
std::vector<torch::Tensor> mlp_chain_bwd(torch::Tensor dy,
                                         std::vector<torch::Tensor> ws,
                                         std::vector<torch::Tensor> acts,
                                         bool need_dx) {
  check_cuda_contig(dy, "dy");
  TORCH_CHECK(dy.scalar_type() == torch::kBFloat16, "dy must be bf16");
  const int L = (int)ws.size();
  const long R = dy.numel() / dy.size(-1);
  int d[9];
  const float* wp[8];
  const void* ap[8] = {nullptr};
  void* zp[8];
  float* dbp[8];
  for (int i = 0; i < L; ++i) {
    wp[i] = ws[i].data_ptr<float>();
    d[i] = (int)ws[i].size(1);
    d[i + 1] = (int)ws[i].size(0);
    if (i < L - 1) ap[i] = acts[i].data_ptr();
  }
  std::vector<torch::Tensor> dzs, dbs;
  for (int i = 0; i < L; ++i) {
    dzs.push_back(torch::empty({R, (long)d[i + 1]},
                               dy.options().dtype(torch::kBFloat16)));
    zp[i] = dzs.back().data_ptr();
    dbs.push_back(torch::zeros({(long)d[i + 1]},
                               dy.options().dtype(torch::kFloat32)));
    dbp[i] = dbs.back().data_ptr<float>();
  }
  torch::Tensor dx;
  if (need_dx)
    dx = torch::empty({R, (long)d[0]}, dy.options().dtype(torch::kBFloat16));
  auto stream = at::hip::getCurrentHIPStream();
  npf_mlp_bwd_launch(dy.data_ptr(), wp, ap, zp,
                     need_dx ? dx.data_ptr() : nullptr, dbp, d, L, R, stream);
  std::vector<torch::Tensor> out = {need_dx ? dx : torch::Tensor()};
  for (auto& z : dzs) out.push_back(z);
  for (auto& b : dbs) out.push_back(b);
  return out;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("attn_fwd", &attn_fwd, "fused cross-attention forward (out, lse)");
  m.def("attn_bwd", &attn_bwd, "fused cross-attention backward (dq, dk, dv)");
  m.def("setconv_fwd", &setconv_fwd, "fused Gaussian SetConv forward");
  m.def("setconv_bwd", &setconv_bwd, "fused Gaussian SetConv backward");
  m.def("gauss_ll_fwd", &gauss_ll_fwd, "fused Gaussian log-lik forward");
  m.def("gauss_ll_bwd", &gauss_ll_bwd, "fused Gaussian log-lik backward");
  m.def("lse_z_fwd", &lse_z_fwd, "NPML logmeanexp-over-z forward");
  m.def("lse_z_bwd", &lse_z_bwd, "NPML logmeanexp-over-z backward");
  m.def("qkv_fwd", &qkv_fwd, "fused K/Q/V projections + head split");
  m.def("qkv_bwd", &qkv_bwd, "fused K/Q/V projection backward");
  m.def("add_ln_fwd", &add_ln_fwd, "fused add + LayerNorm forward");
  m.def("add_ln_bwd", &add_ln_bwd, "fused add + LayerNorm backward");
  m.def("convblock_stats", &convblock_stats,
        "fused conv block: per-channel batch stats (+running update)");
  m.def("convblock_fwd", &convblock_fwd,
        "fused bn+relu+depthwise-conv(+residual) forward, 1D");
  m.def("convblock_bwd", &convblock_bwd,
        "fused conv block backward -> (dx, dw, db, dgamma, dbeta)");
  m.def("convblock2d_fwd", &convblock2d_fwd,
        "fused bn+relu+depthwise-conv(+residual) forward, 2D");
  m.def("convblock2d_bwd", &convblock2d_bwd,
        "fused 2D conv block backward -> (dx, dw, db, dgamma, dbeta)");
  m.def("griddensity_fwd", &griddensity_fwd,
        "fused masked abs-conv density encoder forward");
  m.def("griddensity_bwd", &griddensity_bwd,
        "fused density encoder backward -> (dx, dw)");
  m.def("gauss_kl_fwd", &gauss_kl_fwd, "fused diagonal-Gaussian KL + reduce");
  m.def("gauss_kl_bwd", &gauss_kl_bwd,
        "fused KL backward -> (dmq, dsq, dmp, dsp)");
  m.def("mlp_chain_fwd", &mlp_chain_fwd,
        "fused MFMA MLP chain forward -> [y, act_0..act_{L-2}]");
  m.def("mlp_chain_bwd", &mlp_chain_bwd,
        "fused MLP chain backward -> [dx, dz_0.., db_0..]");
  m.attr("_arch") = "gfx950";
}
