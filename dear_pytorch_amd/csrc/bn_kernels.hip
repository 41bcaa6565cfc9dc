// Fused NHWC BatchNorm(+ReLU)(+residual-add) training kernels for CDNA4.
//
// Replaces the MIOpen BN Spatial kernel stack + separate ReLU / residual-add
// elementwise kernels (~18% of a ResNet-50 fp32 NHWC iteration, see
// profiles/README.md) with:
//   fwd: bn_fwd_stats (partial per-channel sum/sumsq, deterministic 2-stage)
//        bn_fwd_reduce (mean/invstd + running-stat update)
//        bn_fwd_apply (normalize + affine + residual + ReLU, one pass)
//   bwd: bn_bwd_stats (dy_eff = relu-masked dy; partial Σdy, Σdy·x̂;
//        also materializes dy_eff == d_residual for free)
//        bn_bwd_reduce (dgamma/dbeta)
//        bn_bwd_dx (one pass)
//
// Layout contract: x is channels_last (NHWC): a [rows = N*H*W, C] matrix with
// stride C — lane = channel gives perfectly coalesced rows.  fp32, matches
// torch.nn.BatchNorm2d numerics (biased var for normalization, unbiased for
// running stats) to reduction-order tolerance.
#include <hip/hip_runtime.h>

namespace {

constexpr int kThreads = 256;

// thread t covers channel c = blockIdx.y*kThreads + t; rows strided over
// gridDim.x.  Partials: [gridDim.x][C] pairs.
__global__ __launch_bounds__(kThreads) void bn_fwd_stats_kernel(
    const float* __restrict__ x, long rows, int C,
    float* __restrict__ psum, float* __restrict__ psumsq) {
  const int c = blockIdx.y * kThreads + threadIdx.x;
  if (c >= C) return;
  float s = 0.f, ss = 0.f;
  for (long r = blockIdx.x; r < rows; r += gridDim.x) {
    float v = x[r * C + c];
    s += v;
    ss = fmaf(v, v, ss);
  }
  psum[(long)blockIdx.x * C + c] = s;
  psumsq[(long)blockIdx.x * C + c] = ss;
}

__global__ __launch_bounds__(kThreads) void bn_fwd_reduce_kernel(
    const float* __restrict__ psum, const float* __restrict__ psumsq,
    int nparts, int C, long rows, float eps, float momentum,
    float* __restrict__ mean, float* __restrict__ invstd,
    float* __restrict__ running_mean, float* __restrict__ running_var,
    int update_running) {
  const int c = blockIdx.x * kThreads + threadIdx.x;
  if (c >= C) return;
  float s = 0.f, ss = 0.f;
  for (int p = 0; p < nparts; ++p) {
    s += psum[(long)p * C + c];
    ss += psumsq[(long)p * C + c];
  }
  const float m = s / (float)rows;
  float var = fmaxf(ss / (float)rows - m * m, 0.f);
  mean[c] = m;
  invstd[c] = rsqrtf(var + eps);
  if (update_running) {
    const float unbiased = rows > 1 ? var * (float)rows / (float)(rows - 1)
                                    : var;
    running_mean[c] = fmaf(momentum, m - running_mean[c], running_mean[c]);
    running_var[c] = fmaf(momentum, unbiased - running_var[c],
                          running_var[c]);
  }
}

template <bool kRelu, bool kRes>
__global__ __launch_bounds__(kThreads) void bn_fwd_apply_kernel(
    const float* __restrict__ x, const float* __restrict__ res,
    float* __restrict__ y, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ w,
    const float* __restrict__ b, long rows, int C) {
  const int c = blockIdx.y * kThreads + threadIdx.x;
  if (c >= C) return;
  const float mc = mean[c], ic = invstd[c], wc = w[c], bc = b[c];
  const float scale = ic * wc;
  const float shift = fmaf(-mc, scale, bc);
  for (long r = blockIdx.x; r < rows; r += gridDim.x) {
    float v = fmaf(x[r * C + c], scale, shift);
    if (kRes) v += res[r * C + c];
    if (kRelu) v = fmaxf(v, 0.f);
    y[r * C + c] = v;
  }
}

// dy_eff = relu ? dy * (y > 0) : dy ; partials of Σdy_eff and Σdy_eff·x̂.
// Writes dy_eff (this IS the residual grad when the add was fused).
template <bool kRelu, bool kStoreDyEff>
__global__ __launch_bounds__(kThreads) void bn_bwd_stats_kernel(
    const float* __restrict__ x, const float* __restrict__ dy,
    const float* __restrict__ y, float* __restrict__ dy_eff,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    long rows, int C, float* __restrict__ pdb, float* __restrict__ pdg) {
  const int c = blockIdx.y * kThreads + threadIdx.x;
  if (c >= C) return;
  const float mc = mean[c], ic = invstd[c];
  float sdb = 0.f, sdg = 0.f;
  for (long r = blockIdx.x; r < rows; r += gridDim.x) {
    float g = dy[r * C + c];
    if (kRelu) g = y[r * C + c] > 0.f ? g : 0.f;
    if (kStoreDyEff) dy_eff[r * C + c] = g;
    sdb += g;
    sdg = fmaf(g, (x[r * C + c] - mc) * ic, sdg);
  }
  pdb[(long)blockIdx.x * C + c] = sdb;
  pdg[(long)blockIdx.x * C + c] = sdg;
}

__global__ __launch_bounds__(kThreads) void bn_bwd_reduce_kernel(
    const float* __restrict__ pdb, const float* __restrict__ pdg, int nparts,
    int C, float* __restrict__ dbeta, float* __restrict__ dgamma) {
  const int c = blockIdx.x * kThreads + threadIdx.x;
  if (c >= C) return;
  float db = 0.f, dg = 0.f;
  for (int p = 0; p < nparts; ++p) {
    db += pdb[(long)p * C + c];
    dg += pdg[(long)p * C + c];
  }
  dbeta[c] = db;
  dgamma[c] = dg;
}

// dx = w*invstd * (dy_eff - dbeta/M - x̂ * dgamma/M)
template <bool kRelu, bool kHaveDyEff>
__global__ __launch_bounds__(kThreads) void bn_bwd_dx_kernel(
    const float* __restrict__ x, const float* __restrict__ dy,
    const float* __restrict__ y, const float* __restrict__ dy_eff,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ w, const float* __restrict__ dbeta,
    const float* __restrict__ dgamma, long rows, int C,
    float* __restrict__ dx) {
  const int c = blockIdx.y * kThreads + threadIdx.x;
  if (c >= C) return;
  const float mc = mean[c], ic = invstd[c];
  const float k = ic * w[c];
  const float mdb = dbeta[c] / (float)rows;
  const float mdg = dgamma[c] / (float)rows;
  for (long r = blockIdx.x; r < rows; r += gridDim.x) {
    float g;
    if (kHaveDyEff) {
      g = dy_eff[r * C + c];
    } else {
      g = dy[r * C + c];
      if (kRelu) g = y[r * C + c] > 0.f ? g : 0.f;
    }
    const float xh = (x[r * C + c] - mc) * ic;
    dx[r * C + c] = k * (g - mdb - xh * mdg);
  }
}

inline int row_blocks(long rows, int cblocks) {
  // enough workgroups to fill 256 CUs across (row x channel) grid
  long target = 2048 / (cblocks > 0 ? cblocks : 1);
  if (target < 1) target = 1;
  if (target > rows) target = rows;
  if (target > 1024) target = 1024;
  return (int)target;
}

}  // namespace

extern "C" {

void dear_bn_fwd(hipStream_t stream, const float* x, const float* res,
                 float* y, const float* w, const float* b, float* mean,
                 float* invstd, float* running_mean, float* running_var,
                 float* psum, float* psumsq, int nparts, long rows, int C,
                 float eps, float momentum, int training, int relu) {
  const int cblocks = (C + kThreads - 1) / kThreads;
  const int rb = nparts;  // caller sized the partial buffer
  if (training) {
    hipLaunchKernelGGL(bn_fwd_stats_kernel, dim3(rb, cblocks), dim3(kThreads),
                       0, stream, x, rows, C, psum, psumsq);
    hipLaunchKernelGGL(bn_fwd_reduce_kernel, dim3(cblocks), dim3(kThreads), 0,
                       stream, psum, psumsq, rb, C, rows, eps, momentum, mean,
                       invstd, running_mean, running_var, 1);
  }
  const int arb = row_blocks(rows, cblocks);
#define APPLY(R, S)                                                        \
  hipLaunchKernelGGL((bn_fwd_apply_kernel<R, S>), dim3(arb, cblocks),      \
                     dim3(kThreads), 0, stream, x, res, y, mean, invstd, w, \
                     b, rows, C)
  if (relu && res) APPLY(true, true);
  else if (relu) APPLY(true, false);
  else if (res) APPLY(false, true);
  else APPLY(false, false);
#undef APPLY
}

void dear_bn_bwd(hipStream_t stream, const float* x, const float* dy,
                 const float* y, float* dy_eff, const float* w,
                 const float* mean, const float* invstd, float* pdb,
                 float* pdg, int nparts, float* dbeta, float* dgamma,
                 float* dx, long rows, int C, int relu, int want_dy_eff) {
  const int cblocks = (C + kThreads - 1) / kThreads;
  const int rb = nparts;
#define STATS(R, S)                                                         \
  hipLaunchKernelGGL((bn_bwd_stats_kernel<R, S>), dim3(rb, cblocks),        \
                     dim3(kThreads), 0, stream, x, dy, y, dy_eff, mean,     \
                     invstd, rows, C, pdb, pdg)
  if (relu && want_dy_eff) STATS(true, true);
  else if (relu) STATS(true, false);
  else if (want_dy_eff) STATS(false, true);
  else STATS(false, false);
#undef STATS
  hipLaunchKernelGGL(bn_bwd_reduce_kernel, dim3(cblocks), dim3(kThreads), 0,
                     stream, pdb, pdg, rb, C, dbeta, dgamma);
  const int arb = row_blocks(rows, cblocks);
#define DX(R, H)                                                            \
  hipLaunchKernelGGL((bn_bwd_dx_kernel<R, H>), dim3(arb, cblocks),          \
                     dim3(kThreads), 0, stream, x, dy, y, dy_eff, mean,     \
                     invstd, w, dbeta, dgamma, rows, C, dx)
  if (want_dy_eff) DX(false, true);   // dy_eff already materialized
  else if (relu) DX(true, false);
  else DX(false, false);
#undef DX
}

}  // extern "C"
