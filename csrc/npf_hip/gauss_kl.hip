// Fused diagonal-Gaussian KL divergence + latent-set reduction.
//
//   out[b] = sum_m [ log(sp/sq) + (sq^2 + (mq-mp)^2) / (2 sp^2) - 1/2 ]
//
// The NPVI ELBO's KL( q(z|C,T) || q(z|C) ) term (reference losses.py:135-150
// composes torch.distributions.kl_divergence — ~10 elementwise kernels —
// with a view+sum); here it is one kernel forward and one backward.
// One workgroup per batch row, grid-stride over the flattened latent dims.

#include "common.h"

#define KL_BLOCK 256

extern "C" __global__ void __launch_bounds__(KL_BLOCK)
npf_gauss_kl_fwd(const float* __restrict__ mq, const float* __restrict__ sq,
                 const float* __restrict__ mp, const float* __restrict__ sp,
                 float* __restrict__ out, long long rows, long long m) {
  __shared__ float red[16];
  const long long row = blockIdx.x;
  if (row >= rows) return;
  const long long base = row * m;
  float acc = 0.f;
  for (long long i = threadIdx.x; i < m; i += KL_BLOCK) {
    const float dm = mq[base + i] - mp[base + i];
    const float q = sq[base + i], p = sp[base + i];
    const float r = q / p;
    acc += __logf(p) - __logf(q) + 0.5f * (r * r + dm * dm / (p * p)) - 0.5f;
  }
  acc = block_reduce_sum(acc, red);
  if (threadIdx.x == 0) out[row] = acc;
}

extern "C" __global__ void __launch_bounds__(KL_BLOCK)
npf_gauss_kl_bwd(const float* __restrict__ mq, const float* __restrict__ sq,
                 const float* __restrict__ mp, const float* __restrict__ sp,
                 const float* __restrict__ dout, float* __restrict__ dmq,
                 float* __restrict__ dsq, float* __restrict__ dmp,
                 float* __restrict__ dsp, long long rows, long long m) {
  const long long row = blockIdx.x;
  if (row >= rows) return;
  const float g = dout[row];
  const long long base = row * m;
  for (long long i = threadIdx.x; i < m; i += KL_BLOCK) {
    const float dm = mq[base + i] - mp[base + i];
    const float q = sq[base + i], p = sp[base + i];
    const float inv_p2 = 1.f / (p * p);
    dmq[base + i] = g * dm * inv_p2;
    dmp[base + i] = -g * dm * inv_p2;
    dsq[base + i] = g * (q * inv_p2 - 1.f / q);
    dsp[base + i] = g * (1.f / p - (q * q + dm * dm) * inv_p2 / p);
  }
}

extern "C" void npf_gauss_kl_fwd_launch(const float* mq, const float* sq,
                                        const float* mp, const float* sp,
                                        float* out, long long rows,
                                        long long m, hipStream_t stream) {
  hipLaunchKernelGGL(npf_gauss_kl_fwd, dim3((unsigned)rows), dim3(KL_BLOCK), 0,
                     stream, mq, sq, mp, sp, out, rows, m);
}

extern "C" void npf_gauss_kl_bwd_launch(const float* mq, const float* sq,
                                        const float* mp, const float* sp,
                                        const float* dout, float* dmq,
                                        float* dsq, float* dmp, float* dsp,
                                        long long rows, long long m,
                                        hipStream_t stream) {
  hipLaunchKernelGGL(npf_gauss_kl_bwd, dim3((unsigned)rows), dim3(KL_BLOCK), 0,
                     stream, mq, sq, mp, sp, dout, dmq, dsq, dmp, dsp, rows,
                     m);
}
