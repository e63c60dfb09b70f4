// Fused MLP-chain on MFMA (bf16 inputs, fp32 accumulate) — the NPF
// encoder/decoder hot path (SURVEY.md §2.3 rows "MLP decoder" / "DeepSets
// encoder": 4-hidden 128-wide MLPs applied to [Z*B*T, 128] flattened rows).
//
// One kernel runs the WHOLE chain y = W_L(relu(...relu(W_1 x + b_1)...)) + b_L:
// per workgroup a 32-row tile of X stays resident in LDS, each layer's fp32
// master weights are staged (and converted to bf16) into LDS, MFMA 16x16x32
// tiles accumulate in fp32, bias+ReLU happen in registers, and the post-ReLU
// activations stream to HBM only as the backward's saved tensors.
//
// Why: at these shapes (R ~ 5.7k rows, 128 wide) each hipBLASLt GEMM launch
// fills ~64 workgroups for ~35us, and the profiled AttnCNP step spends 41%
// of GPU time in cast/relu elementwise kernels.  The chain kernel removes
// every intermediate HBM round-trip, every weight-cast kernel (fp32->bf16
// conversion rides the LDS staging), and L-1 launches.
//
// Backward: npf_mlp_bwd walks the chain in reverse in one kernel (dz staged
// in LDS, W staged TRANSPOSED, relu mask from the saved activations),
// storing each dz_l to HBM and atomics-reducing db_l; the dW_l rank-k GEMMs
// ([d,R]x[R,d], K large = good library shapes) stay on hipBLASLt, driven
// from the python wrapper.
//
// MFMA fragment layout (cdna_hip_programming.md "Fragment layout", gfx950
// v_mfma_f32_16x16x32_bf16):
//   A: lane holds row = lane&15, k = (lane>>4)*8 .. +8   (8 bf16 = 4 VGPRs)
//   B: lane holds col = lane&15, k = (lane>>4)*8 .. +8
//   C/D: lane holds col = lane&15, row = (lane>>4)*4 + reg (4 fp32)
// Padded dims round up to 32: the MFMA consumes a full K=32 slice per
// issue, so staging must zero-fill the whole 32-wide contraction chunk.
// For y = x @ W^T with torch Linear W[out,in], the B fragment (col=out,
// k=in) reads contiguous rows of W (no transpose); the backward's
// da = dz @ W stages W transposed.
//
// All pointers ride in a by-value kernel-argument struct — no device-side
// pointer arrays, no per-call H2D copies, hipGraph-capture safe.

#include "common.h"

#define MC_MAX_L 8
#define MC_MAX_D 128         // max layer width (padded to 16 internally)
#define MC_TR 32             // rows per workgroup (2 waves x 16)
#define MC_BLOCK 128         // 2 waves
#define MC_PAD 8             // bf16 row padding: 16B-aligned, conflict-free
#define MC_STRIDE (MC_MAX_D + MC_PAD)

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

struct MlpParams {
  const __hip_bfloat16* x;          // [R, d0]
  __hip_bfloat16* y;                // [R, dL]
  const float* w[MC_MAX_L];         // [d_{l+1}, d_l] fp32 master weights
  const float* b[MC_MAX_L];         // [d_{l+1}]
  __hip_bfloat16* act[MC_MAX_L];    // fwd: saved post-relu (l = 0..L-2)
  __hip_bfloat16* dz[MC_MAX_L];     // bwd: stored dz_l
  const __hip_bfloat16* dy;         // [R, dL]
  __hip_bfloat16* dx;               // [R, d0] or null
  float* db[MC_MAX_L];              // [d_{l+1}] fp32 (pre-zeroed)
  int d[MC_MAX_L + 1];
  int L;
  long R;
};

__device__ __forceinline__ bf16x8 ld_frag16(const __hip_bfloat16* p) {
  return *reinterpret_cast<const bf16x8*>(p);  // 16B-aligned by layout
}


// stage a [MC_TR, d] bf16 global tile (rows r0..r0+MC_TR of [R, d]) into
// LDS, zero-padding cols to d_p and rows beyond R; 16B loads when d % 8 == 0
__device__ __forceinline__ void stage_x(
    const __hip_bfloat16* __restrict__ src, __hip_bfloat16 (*dst)[MC_STRIDE],
    long r0, long R, int d, int d_p) {
  const int tr = (int)((R - r0) < MC_TR ? (R - r0) : MC_TR);
  if ((d & 7) == 0) {
    const int qn = d >> 3;
    for (int i = threadIdx.x; i < tr * qn; i += MC_BLOCK) {
      const int r = i / qn, k = (i % qn) << 3;
      const bf16x8 v = *reinterpret_cast<const bf16x8*>(
          &src[(r0 + r) * d + k]);
      *reinterpret_cast<bf16x8*>(&dst[r][k]) = v;
    }
  } else {
    for (int i = threadIdx.x; i < tr * d; i += MC_BLOCK) {
      const int r = i / d, k = i % d;
      dst[r][k] = src[(r0 + r) * d + k];
    }
  }
  const __hip_bfloat16 z = __float2bfloat16(0.f);
  const int kpad = d_p - d;
  if (kpad > 0)
    for (int i = threadIdx.x; i < MC_TR * kpad; i += MC_BLOCK)
      dst[i / kpad][d + i % kpad] = z;
  for (int i = threadIdx.x; i < (MC_TR - tr) * d; i += MC_BLOCK)
    dst[tr + i / d][i % d] = z;
}

// stage a [dout, din] fp32 weight into LDS bf16 (optionally transposed),
// zero-padding up to (dout_p, din_p).  The dense region loads float4
// (4x fewer global round-trips — scalar staging measured 105us/call with
// only ~178 WGs to hide the latency); padding cells are filled separately.
__device__ __forceinline__ void stage_w(
    const float* __restrict__ w, __hip_bfloat16 (*ws)[MC_STRIDE], int dout,
    int din, int dout_p, int din_p, bool transpose) {
  if ((din & 3) == 0) {
    const int qn = din >> 2;
    for (int i = threadIdx.x; i < dout * qn; i += MC_BLOCK) {
      const int o = i / qn, k = (i % qn) << 2;
      const float4 v = reinterpret_cast<const float4*>(w)[i];
      if (transpose) {
        ws[k][o] = __float2bfloat16(v.x);
        ws[k + 1][o] = __float2bfloat16(v.y);
        ws[k + 2][o] = __float2bfloat16(v.z);
        ws[k + 3][o] = __float2bfloat16(v.w);
      } else {
        __hip_bfloat162 p0, p1;
        p0 = __hip_bfloat162{__float2bfloat16(v.x), __float2bfloat16(v.y)};
        p1 = __hip_bfloat162{__float2bfloat16(v.z), __float2bfloat16(v.w)};
        *reinterpret_cast<__hip_bfloat162*>(&ws[o][k]) = p0;
        *reinterpret_cast<__hip_bfloat162*>(&ws[o][k + 2]) = p1;
      }
    }
  } else {
    for (int i = threadIdx.x; i < dout * din; i += MC_BLOCK) {
      const int o = i / din, k = i % din;
      const __hip_bfloat16 hv = __float2bfloat16(w[i]);
      if (transpose)
        ws[k][o] = hv;
      else
        ws[o][k] = hv;
    }
  }
  // zero the padded cells (column pad for every padded row, then row pad)
  const __hip_bfloat16 z = __float2bfloat16(0.f);
  const int kpad = din_p - din;
  if (kpad > 0)
    for (int i = threadIdx.x; i < dout_p * kpad; i += MC_BLOCK) {
      const int o = i / kpad, k = din + i % kpad;
      if (transpose)
        ws[k][o] = z;
      else
        ws[o][k] = z;
    }
  for (int i = threadIdx.x; i < (dout_p - dout) * din; i += MC_BLOCK) {
    const int o = dout + i / din, k = i % din;
    if (transpose)
      ws[k][o] = z;
    else
      ws[o][k] = z;
  }
}

// one [MC_TR, din_p] @ [din_p, dout_p]^T MFMA pass from LDS operands;
// acc covers the calling wave's 16 rows
__device__ __forceinline__ void tile_gemm(
    const __hip_bfloat16 (*a)[MC_STRIDE], const __hip_bfloat16 (*w)[MC_STRIDE],
    f32x4* acc, int din_p, int dout_p, int row0, int lane) {
  for (int kk = 0; kk < din_p; kk += 32) {
    const bf16x8 av = ld_frag16(&a[row0 + (lane & 15)][kk + (lane >> 4) * 8]);
    for (int n = 0; n < dout_p / 16; ++n) {
      const bf16x8 bv = ld_frag16(&w[n * 16 + (lane & 15)][kk + (lane >> 4) * 8]);
      acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av, bv, acc[n], 0, 0, 0);
    }
  }
}

// ---------------------------------------------------------------------------
// forward
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(MC_BLOCK)
npf_mlp_fwd(MlpParams p) {
  __shared__ __align__(16) __hip_bfloat16 a_lds[2][MC_TR][MC_STRIDE];
  __shared__ __align__(16) __hip_bfloat16 w_lds[MC_MAX_D][MC_STRIDE];
  __shared__ float b_lds[MC_MAX_D];

  const long r0 = (long)blockIdx.x * MC_TR;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;

  const int d0 = p.d[0];
  const int d0p = (d0 + 31) & ~31;
  stage_x(p.x, a_lds[0], r0, p.R, d0, d0p);

  int cur = 0;
  for (int l = 0; l < p.L; ++l) {
    const int din = p.d[l], dout = p.d[l + 1];
    const int din_p = (din + 31) & ~31, dout_p = (dout + 31) & ~31;
    stage_w(p.w[l], w_lds, dout, din, dout_p, din_p, false);
    for (int i = threadIdx.x; i < dout_p; i += MC_BLOCK)
      b_lds[i] = (i < dout) ? p.b[l][i] : 0.f;
    __syncthreads();

    const int row0 = wave * 16;
    f32x4 acc[MC_MAX_D / 16];
    #pragma unroll
    for (int n = 0; n < MC_MAX_D / 16; ++n) acc[n] = f32x4{0.f, 0.f, 0.f, 0.f};
    tile_gemm(a_lds[cur], w_lds, acc, din_p, dout_p, row0, lane);
    __syncthreads();  // a_lds[cur] / w_lds free

    const bool is_last = (l == p.L - 1);
    const int nxt = cur ^ 1;
    const int col = lane & 15;
    const int rbase = row0 + (lane >> 4) * 4;
    for (int n = 0; n < dout_p / 16; ++n) {
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int r = rbase + i;
        const int c = n * 16 + col;
        float v = acc[n][i] + b_lds[c];
        if (!is_last) v = fmaxf(v, 0.f);
        const __hip_bfloat16 hv = __float2bfloat16(v);
        a_lds[nxt][r][c] = hv;
        if (r0 + r < p.R && c < dout) {
          if (is_last)
            p.y[(r0 + r) * dout + c] = hv;
          else
            p.act[l][(r0 + r) * dout + c] = hv;
        }
      }
    }
    cur = nxt;
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// backward: dz chain + db atomics; dz_l stored for the library dW GEMMs
// ---------------------------------------------------------------------------

__device__ __forceinline__ void reduce_db(
    const __hip_bfloat16 (*dz)[MC_STRIDE], float* db, int d, long r0, long R) {
  // thread t owns column t (+ strides): 32 LDS reads, one atomic
  const int tr = (int)min((long)MC_TR, R - r0);
  for (int c = threadIdx.x; c < d; c += MC_BLOCK) {
    float s = 0.f;
    for (int r = 0; r < tr; ++r) s += __bfloat162float(dz[r][c]);
    atomicAdd(&db[c], s);
  }
}

extern "C" __global__ void __launch_bounds__(MC_BLOCK)
npf_mlp_bwd(MlpParams p) {
  __shared__ __align__(16) __hip_bfloat16 a_lds[2][MC_TR][MC_STRIDE];
  __shared__ __align__(16) __hip_bfloat16 w_lds[MC_MAX_D][MC_STRIDE];

  const long r0 = (long)blockIdx.x * MC_TR;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int L = p.L;

  // stage dz_{L-1} = dY; store it and its db
  const int dl = p.d[L];
  const int dlp = (dl + 31) & ~31;
  stage_x(p.dy, a_lds[0], r0, p.R, dl, dlp);
  __syncthreads();
  for (int i = threadIdx.x; i < MC_TR * dl; i += MC_BLOCK) {
    const int r = i / dl, c = i % dl;
    if (r0 + r < p.R) p.dz[L - 1][(r0 + r) * dl + c] = a_lds[0][r][c];
  }
  reduce_db(a_lds[0], p.db[L - 1], dl, r0, p.R);

  int cur = 0;
  for (int l = L - 1; l >= 1; --l) {
    const int din = p.d[l], dout = p.d[l + 1];
    const int din_p = (din + 31) & ~31, dout_p = (dout + 31) & ~31;
    // da_{l-1} = dz_l @ W_l : stage W transposed so B[k=out][n=in]
    stage_w(p.w[l], w_lds, dout, din, dout_p, din_p, true);
    __syncthreads();

    const int row0 = wave * 16;
    f32x4 acc[MC_MAX_D / 16];
    #pragma unroll
    for (int n = 0; n < MC_MAX_D / 16; ++n) acc[n] = f32x4{0.f, 0.f, 0.f, 0.f};
    tile_gemm(a_lds[cur], w_lds, acc, dout_p, din_p, row0, lane);
    __syncthreads();

    // dz_{l-1} = da ⊙ relu'(a_{l-1})
    const int nxt = cur ^ 1;
    const int col = lane & 15;
    const int rbase = row0 + (lane >> 4) * 4;
    for (int n = 0; n < din_p / 16; ++n) {
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int r = rbase + i;
        const int c = n * 16 + col;
        __hip_bfloat16 hv = __float2bfloat16(0.f);
        if (r0 + r < p.R && c < din) {
          const float av = __bfloat162float(p.act[l - 1][(r0 + r) * din + c]);
          const float v = (av > 0.f) ? acc[n][i] : 0.f;
          hv = __float2bfloat16(v);
          p.dz[l - 1][(r0 + r) * din + c] = hv;
        }
        a_lds[nxt][r][c] = hv;
      }
    }
    cur = nxt;
    __syncthreads();
    reduce_db(a_lds[cur], p.db[l - 1], din, r0, p.R);
    __syncthreads();
  }

  // dX = dz_0 @ W_0
  if (p.dx != nullptr) {
    const int din = p.d[0], dout = p.d[1];
    const int din_p = (din + 31) & ~31, dout_p = (dout + 31) & ~31;
    stage_w(p.w[0], w_lds, dout, din, dout_p, din_p, true);
    __syncthreads();
    const int row0 = wave * 16;
    f32x4 acc[MC_MAX_D / 16];
    #pragma unroll
    for (int n = 0; n < MC_MAX_D / 16; ++n) acc[n] = f32x4{0.f, 0.f, 0.f, 0.f};
    tile_gemm(a_lds[cur], w_lds, acc, dout_p, din_p, row0, lane);
    const int col = lane & 15;
    const int rbase = row0 + (lane >> 4) * 4;
    for (int n = 0; n < din_p / 16; ++n) {
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int r = rbase + i;
        const int c = n * 16 + col;
        if (r0 + r < p.R && c < din)
          p.dx[(r0 + r) * din + c] = __float2bfloat16(acc[n][i]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// host launchers
// ---------------------------------------------------------------------------

extern "C" void npf_mlp_fwd_launch(const void* x, const float* const* ws,
                                   const float* const* bs, void* const* acts,
                                   void* y, const int* d, int L, long R,
                                   hipStream_t stream) {
  MlpParams p = {};
  p.x = (const __hip_bfloat16*)x;
  p.y = (__hip_bfloat16*)y;
  for (int i = 0; i < L; ++i) {
    p.w[i] = ws[i];
    p.b[i] = bs[i];
    if (i < L - 1) p.act[i] = (__hip_bfloat16*)acts[i];
  }
  for (int i = 0; i <= L; ++i) p.d[i] = d[i];
  p.L = L;
  p.R = R;
  const unsigned grid = (unsigned)((R + MC_TR - 1) / MC_TR);
  hipLaunchKernelGGL(npf_mlp_fwd, dim3(grid), dim3(MC_BLOCK), 0, stream, p);
}

extern "C" void npf_mlp_bwd_launch(const void* dy, const float* const* ws,
                                   const void* const* acts, void* const* dzs,
                                   void* dx, float* const* dbs, const int* d,
                                   int L, long R, hipStream_t stream) {
  MlpParams p = {};
  p.dy = (const __hip_bfloat16*)dy;
  p.dx = (__hip_bfloat16*)dx;
  for (int i = 0; i < L; ++i) {
    p.w[i] = ws[i];
    p.dz[i] = (__hip_bfloat16*)dzs[i];
    if (i < L - 1) p.act[i] = (__hip_bfloat16*)acts[i];
    p.db[i] = dbs[i];
  }
  for (int i = 0; i <= L; ++i) p.d[i] = d[i];
  p.L = L;
  p.R = R;
  const unsigned grid = (unsigned)((R + MC_TR - 1) / MC_TR);
  hipLaunchKernelGGL(npf_mlp_bwd, dim3(grid), dim3(MC_BLOCK), 0, stream, p);
}
