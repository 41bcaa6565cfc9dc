// Fused NHWC BatchNorm(+ReLU)(+residual-add) training kernels for CDNA4.
//
// Replaces the MIOpen BN Spatial kernel stack + separate ReLU / residual-add
// elementwise kernels in the ResNet/DenseNet hot path (profiles/README.md).
//
// Geometry (v2 — the v1 scalar channel-per-lane scheme measured 2x slower
// than MIOpen; this version is float4-vectorized and keeps every lane busy):
//   * the tensor is a [rows = N*H*W, C4 = C/4] float4 matrix (NHWC, C % 4);
//   * a 256-thread block is arranged as rpb rows x cpb float4-columns with
//     cpb = min(C4, 256), rpb = 256 / cpb: every lane moves 16 B per access,
//     fully coalesced for any C >= 4, all lanes busy down to C = 16;
//   * per-channel parameters (scale/shift/mean/invstd/...) are loaded ONCE
//     per thread into registers — zero per-row overhead;
//   * reductions: in-register float4 accumulation over the row stripe, LDS
//     combine across the block's rpb row-threads, deterministic two-stage
//     partial buffer (no atomics), final reduce kernel per channel.
// fp32; matches torch.nn.BatchNorm2d numerics (biased var to normalize,
// unbiased for running stats) to reduction-order tolerance.
#include <hip/hip_runtime.h>

namespace {

constexpr int kThreads = 256;

__device__ __forceinline__ float4 operator+(const float4& a, const float4& b) {
  return make_float4(a.x + b.x, a.y + b.y, a.z + b.z, a.w + b.w);
}

struct Geom {
  int cpb;   // float4 columns per block
  int rpb;   // rows in parallel per block
};

__device__ __forceinline__ bool lane_map(const Geom g, int C4, int& c4,
                                         int& rsub) {
  const int t = threadIdx.x;
  c4 = blockIdx.y * g.cpb + t % g.cpb;
  rsub = t / g.cpb;
  return (c4 < C4) && (rsub < g.rpb);
}

// contiguous slab of rows for this block (DRAM-page locality beats a
// whole-grid stride; measured on the fused_sgd chunks at ~5.8 TB/s)
__device__ __forceinline__ void row_range(long rows, int rpb, long& r0,
                                          long& r1) {
  const long per = ((rows + (long)gridDim.x * rpb - 1) /
                    ((long)gridDim.x * rpb)) * rpb;
  r0 = (long)blockIdx.x * per;
  r1 = r0 + per < rows ? r0 + per : rows;
}

// ---- forward statistics: partial per-channel sum / sumsq ------------------
__global__ __launch_bounds__(kThreads) void bn_fwd_stats_kernel(
    const float4* __restrict__ x, long rows, int C4, Geom g,
    float4* __restrict__ psum, float4* __restrict__ psumsq) {
  __shared__ float4 ls[kThreads], lss[kThreads];
  int c4, rsub;
  const bool act = lane_map(g, C4, c4, rsub);
  float4 s = make_float4(0, 0, 0, 0), ss = make_float4(0, 0, 0, 0);
  if (act) {
    long r0, r1;
    row_range(rows, g.rpb, r0, r1);
    long r = r0 + rsub;
    for (; r + g.rpb < r1; r += 2 * g.rpb) {
      float4 v0 = x[r * C4 + c4];
      float4 v1 = x[(r + g.rpb) * C4 + c4];
      s = s + v0;
      ss.x = fmaf(v0.x, v0.x, ss.x); ss.y = fmaf(v0.y, v0.y, ss.y);
      ss.z = fmaf(v0.z, v0.z, ss.z); ss.w = fmaf(v0.w, v0.w, ss.w);
      s = s + v1;
      ss.x = fmaf(v1.x, v1.x, ss.x); ss.y = fmaf(v1.y, v1.y, ss.y);
      ss.z = fmaf(v1.z, v1.z, ss.z); ss.w = fmaf(v1.w, v1.w, ss.w);
    }
    if (r < r1) {
      float4 v = x[r * C4 + c4];
      s = s + v;
      ss.x = fmaf(v.x, v.x, ss.x); ss.y = fmaf(v.y, v.y, ss.y);
      ss.z = fmaf(v.z, v.z, ss.z); ss.w = fmaf(v.w, v.w, ss.w);
    }
  }
  ls[threadIdx.x] = s;
  lss[threadIdx.x] = ss;
  __syncthreads();
  if (rsub == 0 && c4 < C4) {
    const int cl = threadIdx.x;  // rsub==0 => threadIdx.x == c4 local
    for (int k = 1; k < g.rpb; ++k) {
      s = s + ls[k * g.cpb + cl];
      ss = ss + lss[k * g.cpb + cl];
    }
    psum[(long)blockIdx.x * C4 + c4] = s;
    psumsq[(long)blockIdx.x * C4 + c4] = ss;
  }
}

// ---- stage-A column reduce: [nparts][C4] partial pair -> [S][C4] ----------
// The final reduce kernels run ONE block for small C (cblocks==1) and were
// measured latency-bound at ~17 us; this deterministic pre-reduce splits the
// partial dimension over blockIdx.y (chunk rows each) so the heavy part of
// the reduction is parallel, leaving the single-block finalize <= 64 rows.
constexpr int kRedChunk = 64;

__global__ __launch_bounds__(kThreads) void col_reduce_pair_kernel(
    const float4* __restrict__ a, const float4* __restrict__ b, int nparts,
    int C4, int cpb2, float4* __restrict__ oa, float4* __restrict__ ob) {
  __shared__ float4 l1[kThreads], l2[kThreads];
  const int spb = kThreads / cpb2;
  const int cl = threadIdx.x % cpb2;
  const int sl = threadIdx.x / cpb2;
  const int c4 = blockIdx.x * cpb2 + cl;
  const int s0 = blockIdx.y * kRedChunk;
  const int s1 = s0 + kRedChunk < nparts ? s0 + kRedChunk : nparts;
  float4 s = make_float4(0, 0, 0, 0), ss = make_float4(0, 0, 0, 0);
  if (c4 < C4 && sl < spb) {
    for (int p = s0 + sl; p < s1; p += spb) {
      s = s + a[(long)p * C4 + c4];
      ss = ss + b[(long)p * C4 + c4];
    }
  }
  l1[threadIdx.x] = s;
  l2[threadIdx.x] = ss;
  __syncthreads();
  if (sl == 0 && c4 < C4) {
    for (int k = 1; k < spb; ++k) {
      s = s + l1[k * cpb2 + cl];
      ss = ss + l2[k * cpb2 + cl];
    }
    oa[(long)blockIdx.y * C4 + c4] = s;
    ob[(long)blockIdx.y * C4 + c4] = ss;
  }
}

// ---- finalize mean / invstd + running stats -------------------------------
// Cooperative column reduce of the [nparts][C4] float4 partial matrix:
// a block covers cpb2 float4-columns x spb partial-slices (the v2 scalar
// one-thread-per-channel loop measured 179 us -- it was the whole overhead).
__global__ __launch_bounds__(kThreads) void bn_fwd_reduce_kernel(
    const float4* __restrict__ psum, const float4* __restrict__ psumsq,
    int nparts, int C4, long rows, float eps, float momentum,
    float* __restrict__ mean, float* __restrict__ invstd,
    float* __restrict__ running_mean, float* __restrict__ running_var,
    int update_running, int cpb2) {
  __shared__ float4 l1[kThreads], l2[kThreads];
  const int spb = kThreads / cpb2;
  const int cl = threadIdx.x % cpb2;
  const int sl = threadIdx.x / cpb2;
  const int c4 = blockIdx.x * cpb2 + cl;
  float4 s = make_float4(0, 0, 0, 0), ss = make_float4(0, 0, 0, 0);
  if (c4 < C4 && sl < spb) {
    for (int p = sl; p < nparts; p += spb) {
      s = s + psum[(long)p * C4 + c4];
      ss = ss + psumsq[(long)p * C4 + c4];
    }
  }
  l1[threadIdx.x] = s;
  l2[threadIdx.x] = ss;
  __syncthreads();
  if (sl == 0 && c4 < C4) {
    for (int k = 1; k < spb; ++k) {
      s = s + l1[k * cpb2 + cl];
      ss = ss + l2[k * cpb2 + cl];
    }
    const float inv_rows = 1.f / (float)rows;
    const float unb = rows > 1 ? (float)rows / (float)(rows - 1) : 1.f;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int c = c4 * 4 + j;
      const float sj = (&s.x)[j], ssj = (&ss.x)[j];
      const float m = sj * inv_rows;
      float var = fmaxf(ssj * inv_rows - m * m, 0.f);
      mean[c] = m;
      invstd[c] = rsqrtf(var + eps);
      if (update_running) {
        running_mean[c] = fmaf(momentum, m - running_mean[c],
                               running_mean[c]);
        running_var[c] = fmaf(momentum, var * unb - running_var[c],
                              running_var[c]);
      }
    }
  }
}

// ---- forward apply: y = act(x * scale + shift [+ res]) --------------------
template <bool kRelu, bool kRes>
__global__ __launch_bounds__(kThreads) void bn_fwd_apply_kernel(
    const float4* __restrict__ x, const float4* __restrict__ res,
    float4* __restrict__ y, const float4* __restrict__ mean,
    const float4* __restrict__ invstd, const float4* __restrict__ w,
    const float4* __restrict__ b, long rows, int C4, Geom g) {
  int c4, rsub;
  if (!lane_map(g, C4, c4, rsub)) return;
  const float4 mc = mean[c4], ic = invstd[c4], wc = w[c4], bc = b[c4];
  float4 scale, shift;
  scale.x = ic.x * wc.x; shift.x = fmaf(-mc.x, scale.x, bc.x);
  scale.y = ic.y * wc.y; shift.y = fmaf(-mc.y, scale.y, bc.y);
  scale.z = ic.z * wc.z; shift.z = fmaf(-mc.z, scale.z, bc.z);
  scale.w = ic.w * wc.w; shift.w = fmaf(-mc.w, scale.w, bc.w);
  long r0, r1;
  row_range(rows, g.rpb, r0, r1);
  auto body = [&](long i) {
    float4 v = x[i];
    v.x = fmaf(v.x, scale.x, shift.x);
    v.y = fmaf(v.y, scale.y, shift.y);
    v.z = fmaf(v.z, scale.z, shift.z);
    v.w = fmaf(v.w, scale.w, shift.w);
    if (kRes) {
      float4 q = res[i];
      v.x += q.x; v.y += q.y; v.z += q.z; v.w += q.w;
    }
    if (kRelu) {
      v.x = fmaxf(v.x, 0.f); v.y = fmaxf(v.y, 0.f);
      v.z = fmaxf(v.z, 0.f); v.w = fmaxf(v.w, 0.f);
    }
    y[i] = v;
  };
  long r = r0 + rsub;
  for (; r + g.rpb < r1; r += 2 * g.rpb) {
    body(r * C4 + c4);
    body((r + g.rpb) * C4 + c4);
  }
  if (r < r1) body(r * C4 + c4);
}

// ---- backward statistics: dy_eff, partial Σdy_eff and Σdy_eff·x̂ -----------
// kRecomp (relu, no residual): the ReLU mask is recomputed from the
// pre-activation fma(x, scale, shift) — bitwise-identical to the forward's —
// instead of reading y, eliminating one full HBM pass over the tensor here
// and another in bn_bwd_dx (7 -> 5 passes for the no-res ReLU BNs).
template <bool kRelu, bool kStoreDyEff, bool kRecomp>
__global__ __launch_bounds__(kThreads) void bn_bwd_stats_kernel(
    const float4* __restrict__ x, const float4* __restrict__ dy,
    const float4* __restrict__ y, float4* __restrict__ dy_eff,
    const float4* __restrict__ mean, const float4* __restrict__ invstd,
    const float4* __restrict__ w, const float4* __restrict__ bp,
    long rows, int C4, Geom g, float4* __restrict__ pdb,
    float4* __restrict__ pdg) {
  __shared__ float4 ldb[kThreads], ldg[kThreads];
  int c4, rsub;
  const bool act = lane_map(g, C4, c4, rsub);
  float4 sdb = make_float4(0, 0, 0, 0), sdg = make_float4(0, 0, 0, 0);
  if (act) {
    const float4 mc = mean[c4], ic = invstd[c4];
    float4 scale, shift;
    if (kRecomp) {
      const float4 wc = w[c4], bc = bp[c4];
      scale.x = ic.x * wc.x; shift.x = fmaf(-mc.x, scale.x, bc.x);
      scale.y = ic.y * wc.y; shift.y = fmaf(-mc.y, scale.y, bc.y);
      scale.z = ic.z * wc.z; shift.z = fmaf(-mc.z, scale.z, bc.z);
      scale.w = ic.w * wc.w; shift.w = fmaf(-mc.w, scale.w, bc.w);
    }
    long r0, r1;
    row_range(rows, g.rpb, r0, r1);
    auto body = [&](long i) {
      float4 gg = dy[i];
      const float4 xx = x[i];
      if (kRelu) {
        if (kRecomp) {
          gg.x = fmaf(xx.x, scale.x, shift.x) > 0.f ? gg.x : 0.f;
          gg.y = fmaf(xx.y, scale.y, shift.y) > 0.f ? gg.y : 0.f;
          gg.z = fmaf(xx.z, scale.z, shift.z) > 0.f ? gg.z : 0.f;
          gg.w = fmaf(xx.w, scale.w, shift.w) > 0.f ? gg.w : 0.f;
        } else {
          const float4 yy = y[i];
          gg.x = yy.x > 0.f ? gg.x : 0.f;
          gg.y = yy.y > 0.f ? gg.y : 0.f;
          gg.z = yy.z > 0.f ? gg.z : 0.f;
          gg.w = yy.w > 0.f ? gg.w : 0.f;
        }
      }
      if (kStoreDyEff) dy_eff[i] = gg;
      sdb = sdb + gg;
      sdg.x = fmaf(gg.x, (xx.x - mc.x) * ic.x, sdg.x);
      sdg.y = fmaf(gg.y, (xx.y - mc.y) * ic.y, sdg.y);
      sdg.z = fmaf(gg.z, (xx.z - mc.z) * ic.z, sdg.z);
      sdg.w = fmaf(gg.w, (xx.w - mc.w) * ic.w, sdg.w);
    };
    // 2x row unroll (matches bn_fwd_stats): two float4 loads in flight per
    // lane hides HBM latency at the low-occupancy small-spatial shapes
    long r = r0 + rsub;
    for (; r + g.rpb < r1; r += 2 * g.rpb) {
      body(r * C4 + c4);
      body((r + g.rpb) * C4 + c4);
    }
    if (r < r1) body(r * C4 + c4);
  }
  ldb[threadIdx.x] = sdb;
  ldg[threadIdx.x] = sdg;
  __syncthreads();
  if (rsub == 0 && c4 < C4) {
    const int cl = threadIdx.x;
    for (int k = 1; k < g.rpb; ++k) {
      sdb = sdb + ldb[k * g.cpb + cl];
      sdg = sdg + ldg[k * g.cpb + cl];
    }
    pdb[(long)blockIdx.x * C4 + c4] = sdb;
    pdg[(long)blockIdx.x * C4 + c4] = sdg;
  }
}

__global__ __launch_bounds__(kThreads) void bn_bwd_reduce_kernel(
    const float4* __restrict__ pdb, const float4* __restrict__ pdg,
    int nparts, int C4, float* __restrict__ dbeta, float* __restrict__ dgamma,
    int cpb2) {
  __shared__ float4 l1[kThreads], l2[kThreads];
  const int spb = kThreads / cpb2;
  const int cl = threadIdx.x % cpb2;
  const int sl = threadIdx.x / cpb2;
  const int c4 = blockIdx.x * cpb2 + cl;
  float4 db = make_float4(0, 0, 0, 0), dg = make_float4(0, 0, 0, 0);
  if (c4 < C4 && sl < spb) {
    for (int p = sl; p < nparts; p += spb) {
      db = db + pdb[(long)p * C4 + c4];
      dg = dg + pdg[(long)p * C4 + c4];
    }
  }
  l1[threadIdx.x] = db;
  l2[threadIdx.x] = dg;
  __syncthreads();
  if (sl == 0 && c4 < C4) {
    for (int k = 1; k < spb; ++k) {
      db = db + l1[k * cpb2 + cl];
      dg = dg + l2[k * cpb2 + cl];
    }
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      dbeta[c4 * 4 + j] = (&db.x)[j];
      dgamma[c4 * 4 + j] = (&dg.x)[j];
    }
  }
}

// ---- backward dx: dx = w*invstd*(dy_eff - Σdb/M - x̂·Σdg/M) ----------------
template <bool kRelu, bool kHaveDyEff, bool kRecomp>
__global__ __launch_bounds__(kThreads) void bn_bwd_dx_kernel(
    const float4* __restrict__ x, const float4* __restrict__ dy,
    const float4* __restrict__ y, const float4* __restrict__ dy_eff,
    const float4* __restrict__ mean, const float4* __restrict__ invstd,
    const float4* __restrict__ w, const float4* __restrict__ bp,
    const float4* __restrict__ dbeta,
    const float4* __restrict__ dgamma, long rows, int C4, Geom g,
    float4* __restrict__ dx) {
  int c4, rsub;
  if (!lane_map(g, C4, c4, rsub)) return;
  const float4 mc = mean[c4], ic = invstd[c4], wc = w[c4];
  const float inv_m = 1.f / (float)rows;
  const float4 db = dbeta[c4], dg = dgamma[c4];
  float4 k, mdb, mdg, shift;
  k.x = ic.x * wc.x; mdb.x = db.x * inv_m; mdg.x = dg.x * inv_m;
  k.y = ic.y * wc.y; mdb.y = db.y * inv_m; mdg.y = dg.y * inv_m;
  k.z = ic.z * wc.z; mdb.z = db.z * inv_m; mdg.z = dg.z * inv_m;
  k.w = ic.w * wc.w; mdb.w = db.w * inv_m; mdg.w = dg.w * inv_m;
  if (kRecomp) {
    const float4 bc = bp[c4];
    // k == the forward's scale; shift identical to the forward's
    shift.x = fmaf(-mc.x, k.x, bc.x);
    shift.y = fmaf(-mc.y, k.y, bc.y);
    shift.z = fmaf(-mc.z, k.z, bc.z);
    shift.w = fmaf(-mc.w, k.w, bc.w);
  }
  long r0, r1;
  row_range(rows, g.rpb, r0, r1);
  auto body = [&](long i) {
    const float4 xx = x[i];
    float4 gg;
    if (kHaveDyEff) {
      gg = dy_eff[i];
    } else {
      gg = dy[i];
      if (kRelu && kRecomp) {
        gg.x = fmaf(xx.x, k.x, shift.x) > 0.f ? gg.x : 0.f;
        gg.y = fmaf(xx.y, k.y, shift.y) > 0.f ? gg.y : 0.f;
        gg.z = fmaf(xx.z, k.z, shift.z) > 0.f ? gg.z : 0.f;
        gg.w = fmaf(xx.w, k.w, shift.w) > 0.f ? gg.w : 0.f;
      } else if (kRelu) {
        const float4 yy = y[i];
        gg.x = yy.x > 0.f ? gg.x : 0.f;
        gg.y = yy.y > 0.f ? gg.y : 0.f;
        gg.z = yy.z > 0.f ? gg.z : 0.f;
        gg.w = yy.w > 0.f ? gg.w : 0.f;
      }
    }
    float4 o;
    o.x = k.x * (gg.x - mdb.x - (xx.x - mc.x) * ic.x * mdg.x);
    o.y = k.y * (gg.y - mdb.y - (xx.y - mc.y) * ic.y * mdg.y);
    o.z = k.z * (gg.z - mdb.z - (xx.z - mc.z) * ic.z * mdg.z);
    o.w = k.w * (gg.w - mdb.w - (xx.w - mc.w) * ic.w * mdg.w);
    dx[i] = o;
  };
  long r = r0 + rsub;
  for (; r + g.rpb < r1; r += 2 * g.rpb) {  // 2x unroll (see bwd_stats)
    body(r * C4 + c4);
    body((r + g.rpb) * C4 + c4);
  }
  if (r < r1) body(r * C4 + c4);
}

struct LaunchCfg {
  Geom g;
  int rb;        // row blocks for the elementwise kernels (gridDim.x)
  int stats_rb;  // row blocks for the reduction kernels (= #partials)
  int cblocks;   // gridDim.y
};

LaunchCfg make_cfg(long rows, int C4, int nparts_cap) {
  LaunchCfg cfg;
  cfg.g.cpb = C4 < kThreads ? C4 : kThreads;
  cfg.g.rpb = kThreads / cfg.g.cpb;
  cfg.cblocks = (C4 + cfg.g.cpb - 1) / cfg.g.cpb;
  long max_rb = (rows + cfg.g.rpb - 1) / cfg.g.rpb;
  // elementwise kernels: fill the chip, grid-stride the rest
  long target = 2048 / cfg.cblocks;
  if (target < 1) target = 1;
  cfg.rb = (int)(target < max_rb ? target : max_rb);
  if (cfg.rb < 1) cfg.rb = 1;
  // reduction kernels: each block should chew >= 32 row-iterations so the
  // partial buffers stay small vs the tensor (v2 used a fixed 1024 and the
  // partial traffic dominated small layers)
  long srb = (max_rb + 31) / 32;
  long scap = 1024 / cfg.cblocks;
  if (scap < 8) scap = 8;
  if (srb > scap) srb = scap;
  if (srb > max_rb) srb = max_rb;
  if (srb < 1) srb = 1;
  cfg.stats_rb = (int)srb;
  if (nparts_cap > 0 && cfg.stats_rb > nparts_cap)
    cfg.stats_rb = nparts_cap;
  return cfg;
}

}  // namespace

// two-stage reduce helper: pre-reduce [nparts][C4] pairs into the scratch
// rows the python side allocates PAST the nparts partials (no extra buffers,
// no binding change); returns the partial count the finalize kernel should
// read and swaps the input pointers to the scratch when staged.
static int pre_reduce(hipStream_t stream, int nparts, int C4,
                      const float4*& a, const float4*& b, float4* sa,
                      float4* sb) {
  if (nparts <= kRedChunk) return nparts;
  const int S = (nparts + kRedChunk - 1) / kRedChunk;
  const int cpb2 = C4 < 64 ? C4 : 64;
  const int crb = (C4 + cpb2 - 1) / cpb2;
  hipLaunchKernelGGL(col_reduce_pair_kernel, dim3(crb, S), dim3(kThreads), 0,
                     stream, a, b, nparts, C4, cpb2, sa, sb);
  a = sa;
  b = sb;
  return S;
}

extern "C" {

// host-visible helper so the python side can size partial buffers identically
// (stats_rb rows + ceil(stats_rb/kRedChunk) stage-A scratch rows)
int dear_bn_nparts(long rows, int C) {
  const int n = make_cfg(rows, C / 4, 0).stats_rb;
  return n + (n + kRedChunk - 1) / kRedChunk;
}

void dear_bn_fwd(hipStream_t stream, const float* x, const float* res,
                 float* y, const float* w, const float* b, float* mean,
                 float* invstd, float* running_mean, float* running_var,
                 float* psum, float* psumsq, int nparts, long rows, int C,
                 float eps, float momentum, int training, int relu) {
  const int C4 = C / 4;
  LaunchCfg cfg = make_cfg(rows, C4, nparts);
  if (training) {
    hipLaunchKernelGGL(bn_fwd_stats_kernel,
                       dim3(cfg.stats_rb, cfg.cblocks), dim3(kThreads), 0,
                       stream, (const float4*)x, rows, C4, cfg.g,
                       (float4*)psum, (float4*)psumsq);
    const float4* pa = (const float4*)psum;
    const float4* pb = (const float4*)psumsq;
    float4* sa = (float4*)psum + (size_t)cfg.stats_rb * C4;
    float4* sb = (float4*)psumsq + (size_t)cfg.stats_rb * C4;
    const int np = pre_reduce(stream, cfg.stats_rb, C4, pa, pb, sa, sb);
    const int cpb2 = C4 < 64 ? C4 : 64;
    const int crb = (C4 + cpb2 - 1) / cpb2;
    hipLaunchKernelGGL(bn_fwd_reduce_kernel, dim3(crb), dim3(kThreads), 0,
                       stream, pa, pb,
                       np, C4, rows, eps, momentum, mean, invstd,
                       running_mean, running_var, 1, cpb2);
  }
#define APPLY(R, S)                                                           \
  hipLaunchKernelGGL((bn_fwd_apply_kernel<R, S>),                             \
                     dim3(cfg.rb, cfg.cblocks), dim3(kThreads), 0, stream,    \
                     (const float4*)x, (const float4*)res, (float4*)y,        \
                     (const float4*)mean, (const float4*)invstd,              \
                     (const float4*)w, (const float4*)b, rows, C4, cfg.g)
  if (relu && res) APPLY(true, true);
  else if (relu) APPLY(true, false);
  else if (res) APPLY(false, true);
  else APPLY(false, false);
#undef APPLY
}

void dear_bn_bwd(hipStream_t stream, const float* x, const float* dy,
                 const float* y, float* dy_eff, const float* w,
                 const float* b, const float* mean, const float* invstd,
                 float* pdb, float* pdg, int nparts, float* dbeta,
                 float* dgamma, float* dx, long rows, int C, int relu,
                 int want_dy_eff) {
  const int C4 = C / 4;
  LaunchCfg cfg = make_cfg(rows, C4, nparts);
#define STATS(R, S, RC)                                                       \
  hipLaunchKernelGGL((bn_bwd_stats_kernel<R, S, RC>),                         \
                     dim3(cfg.stats_rb, cfg.cblocks), dim3(kThreads), 0,      \
                     stream,                                                  \
                     (const float4*)x, (const float4*)dy, (const float4*)y,   \
                     (float4*)dy_eff, (const float4*)mean,                    \
                     (const float4*)invstd, (const float4*)w,                 \
                     (const float4*)b, rows, C4, cfg.g, (float4*)pdb,         \
                     (float4*)pdg)
  if (relu && want_dy_eff) STATS(true, true, false);
  else if (relu) STATS(true, false, true);   // no-res ReLU: y never read
  else if (want_dy_eff) STATS(false, true, false);
  else STATS(false, false, false);
#undef STATS
  const float4* pa = (const float4*)pdb;
  const float4* pb = (const float4*)pdg;
  float4* sa = (float4*)pdb + (size_t)cfg.stats_rb * C4;
  float4* sb = (float4*)pdg + (size_t)cfg.stats_rb * C4;
  const int np = pre_reduce(stream, cfg.stats_rb, C4, pa, pb, sa, sb);
  const int cpb2 = C4 < 64 ? C4 : 64;
  const int crb = (C4 + cpb2 - 1) / cpb2;
  hipLaunchKernelGGL(bn_bwd_reduce_kernel, dim3(crb), dim3(kThreads), 0,
                     stream, pa, pb,
                     np, C4, dbeta, dgamma, cpb2);
#define DX(R, H, RC)                                                          \
  hipLaunchKernelGGL((bn_bwd_dx_kernel<R, H, RC>), dim3(cfg.rb, cfg.cblocks), \
                     dim3(kThreads), 0, stream, (const float4*)x,             \
                     (const float4*)dy, (const float4*)y,                     \
                     (const float4*)dy_eff, (const float4*)mean,              \
                     (const float4*)invstd, (const float4*)w,                 \
                     (const float4*)b, (const float4*)dbeta,                  \
                     (const float4*)dgamma, rows, C4, cfg.g, (float4*)dx)
  if (want_dy_eff) DX(false, true, false);
  else if (relu) DX(true, false, true);      // no-res ReLU: y never read
  else DX(false, false, false);
#undef DX
}

}  // extern "C"
