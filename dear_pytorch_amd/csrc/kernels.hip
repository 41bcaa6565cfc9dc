// CDNA4 (gfx950) kernels for the DeAR engine — fused multi-tensor optimizer
// updates, bucket pack/unpack, and device top-k.
//
// Replaces the reference's hot ATen sequences (dear/dopt_rsag.py:254-332:
// per-param pad_buffer copies + 5-6 small kernels per param for the inline
// SGD) with one HBM-bandwidth-bound pass per bucket group, and provides the
// native top-k the reference imported from the missing `tcmm` module
// (wfbp/dopt.py:95).
//
// Design notes (see /opt/skills/guides/cdna_hip_programming.md):
//  * wave64; blocks of 256 threads; float4 (dwordx4) vectorized main path
//    with scalar head/tail for unaligned chunk edges;
//  * grids sized >> 256 workgroups via 64 KiB chunk descriptors (one static
//    descriptor table per bucket group, built once at init, resident in HBM);
//  * memory-bound: each fused_sgd pass moves ~4 streams (g, p, m in, p, m, g
//    out) — everything is fused so each byte crosses HBM once;
//  * no dual CUDA paths, no hipify: HIP-only source for gfx950.
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#define HIP_CHECK(cmd)                                                        \
  do {                                                                        \
    hipError_t e_ = (cmd);                                                    \
    if (e_ != hipSuccess)                                                     \
      throw std::runtime_error(std::string("HIP error: ") +                   \
                               hipGetErrorString(e_) + " at " #cmd);          \
  } while (0)

namespace {

constexpr int kBlock = 256;

// Chunk descriptor rows: {param_ptr, bucket_off, n} int64 triples
// (built by ops/fused._build_desc, read directly from the desc array).
__device__ __forceinline__ bool aligned16(const void* p) {
  return (reinterpret_cast<uintptr_t>(p) & 15u) == 0;
}

// ---------------------------------------------------------------- fused SGD
// One pass: d = g*scale (+ maximize) (+ wd*p); m = first ? d : mu*m+(1-damp)d;
// step = nesterov ? d + mu*m : m; p -= lr*step; g = 0.
template <bool kMom, bool kNesterov, bool kFirst>
__global__ __launch_bounds__(kBlock) void fused_sgd_kernel(
    const int64_t* __restrict__ desc, int nchunks, float* __restrict__ bucket,
    float* __restrict__ mom, float lr, float mu, float damp, float wd,
    float scale) {
  const int c = blockIdx.x;
  if (c >= nchunks) return;
  float* __restrict__ p = reinterpret_cast<float*>(desc[3 * c]);
  const int64_t off = desc[3 * c + 1];
  const int n = (int)desc[3 * c + 2];
  float* __restrict__ g = bucket + off;
  float* __restrict__ m = mom + off;

  const bool vec = aligned16(p) && aligned16(g) && aligned16(m);
  if (vec) {
    const int n4 = n >> 2;
    float4* p4 = reinterpret_cast<float4*>(p);
    float4* g4 = reinterpret_cast<float4*>(g);
    float4* m4 = reinterpret_cast<float4*>(m);
    for (int i = threadIdx.x; i < n4; i += kBlock) {
      float4 gv = g4[i];
      float4 pv = p4[i];
      float4 mv;
      if (kMom && !kFirst) mv = m4[i];
      float out[4];
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float gj = (&gv.x)[j] * scale;
        float pj = (&pv.x)[j];
        float d = gj + wd * pj;
        float step;
        if (kMom) {
          float mj = kFirst ? d : (fmaf(mu, (&mv.x)[j], (1.f - damp) * d));
          (&mv.x)[j] = mj;
          step = kNesterov ? fmaf(mu, mj, d) : mj;
        } else {
          step = d;
        }
        out[j] = fmaf(-lr, step, pj);
      }
      p4[i] = make_float4(out[0], out[1], out[2], out[3]);
      if (kMom) m4[i] = mv;
      g4[i] = make_float4(0.f, 0.f, 0.f, 0.f);
    }
    for (int i = (n4 << 2) + threadIdx.x; i < n; i += kBlock) {
      float gj = g[i] * scale, pj = p[i];
      float d = gj + wd * pj, step;
      if (kMom) {
        float mj = kFirst ? d : fmaf(mu, m[i], (1.f - damp) * d);
        m[i] = mj;
        step = kNesterov ? fmaf(mu, mj, d) : mj;
      } else {
        step = d;
      }
      p[i] = fmaf(-lr, step, pj);
      g[i] = 0.f;
    }
  } else {
    for (int i = threadIdx.x; i < n; i += kBlock) {
      float gj = g[i] * scale, pj = p[i];
      float d = gj + wd * pj, step;
      if (kMom) {
        float mj = kFirst ? d : fmaf(mu, m[i], (1.f - damp) * d);
        m[i] = mj;
        step = kNesterov ? fmaf(mu, mj, d) : mj;
      } else {
        step = d;
      }
      p[i] = fmaf(-lr, step, pj);
      g[i] = 0.f;
    }
  }
}

// --------------------------------------------------------------- fused Adam
template <bool kDecoupled>
__global__ __launch_bounds__(kBlock) void fused_adam_kernel(
    const int64_t* __restrict__ desc, int nchunks, float* __restrict__ bucket,
    float* __restrict__ ma, float* __restrict__ va, float lr, float b1,
    float b2, float eps, float wd, float scale, float inv_bc1,
    float inv_sqrt_bc2) {
  const int c = blockIdx.x;
  if (c >= nchunks) return;
  float* __restrict__ p = reinterpret_cast<float*>(desc[3 * c]);
  const int64_t off = desc[3 * c + 1];
  const int n = (int)desc[3 * c + 2];
  float* __restrict__ g = bucket + off;
  float* __restrict__ m = ma + off;
  float* __restrict__ v = va + off;

  auto upd = [&](float gj, float pj, float& mj, float& vj) -> float {
    gj *= scale;
    if (kDecoupled) {
      pj *= (1.f - lr * wd);
    } else if (wd != 0.f) {
      gj = fmaf(wd, pj, gj);
    }
    mj = fmaf(b1, mj, (1.f - b1) * gj);
    vj = fmaf(b2, vj, (1.f - b2) * gj * gj);
    // p -= lr/bc1 * m / (sqrt(v)/sqrt(bc2) + eps)   (torch.optim.Adam order)
    float denom = fmaf(sqrtf(vj), inv_sqrt_bc2, eps);
    return fmaf(-(lr * inv_bc1), mj / denom, pj);
  };

  if (aligned16(p) && aligned16(g) && aligned16(m) && aligned16(v)) {
    const int n4 = n >> 2;
    float4* p4 = reinterpret_cast<float4*>(p);
    float4* g4 = reinterpret_cast<float4*>(g);
    float4* m4 = reinterpret_cast<float4*>(m);
    float4* v4 = reinterpret_cast<float4*>(v);
    for (int i = threadIdx.x; i < n4; i += kBlock) {
      float4 gv = g4[i], pv = p4[i], mv = m4[i], vv = v4[i];
#pragma unroll
      for (int j = 0; j < 4; ++j)
        (&pv.x)[j] = upd((&gv.x)[j], (&pv.x)[j], (&mv.x)[j], (&vv.x)[j]);
      p4[i] = pv;
      m4[i] = mv;
      v4[i] = vv;
      g4[i] = make_float4(0.f, 0.f, 0.f, 0.f);
    }
    for (int i = (n4 << 2) + threadIdx.x; i < n; i += kBlock) {
      float mj = m[i], vj = v[i];
      p[i] = upd(g[i], p[i], mj, vj);
      m[i] = mj;
      v[i] = vj;
      g[i] = 0.f;
    }
  } else {
    for (int i = threadIdx.x; i < n; i += kBlock) {
      float mj = m[i], vj = v[i];
      p[i] = upd(g[i], p[i], mj, vj);
      m[i] = mj;
      v[i] = vj;
      g[i] = 0.f;
    }
  }
}

// --------------------------------------------------- pack / unpack (generic)
// For baselines (WFBP pull-into-grad path) and any non-view use: gather many
// tensor chunks into a flat bucket / scatter back with optional scale.
__global__ __launch_bounds__(kBlock) void pack_kernel(
    const int64_t* __restrict__ desc, int nchunks,
    float* __restrict__ bucket) {
  const int c = blockIdx.x;
  if (c >= nchunks) return;
  const float* __restrict__ src = reinterpret_cast<const float*>(desc[3 * c]);
  float* __restrict__ dst = bucket + desc[3 * c + 1];
  const int n = (int)desc[3 * c + 2];
  if (aligned16(src) && aligned16(dst) && (n & 3) == 0) {
    const float4* s4 = reinterpret_cast<const float4*>(src);
    float4* d4 = reinterpret_cast<float4*>(dst);
    for (int i = threadIdx.x; i < (n >> 2); i += kBlock) d4[i] = s4[i];
  } else {
    for (int i = threadIdx.x; i < n; i += kBlock) dst[i] = src[i];
  }
}

// += variant for packed-grad mode (dear.py): autograd ASSIGNS fresh grad
// tensors (p.grad=None before backward — no per-param CUDAFunctor_add), and
// one pack_add per bucket group folds them into the pre-zeroed bucket.
// Accumulates so multi-micro-batch gradient accumulation still sums.
__global__ __launch_bounds__(kBlock) void pack_add_kernel(
    const int64_t* __restrict__ desc, int nchunks,
    float* __restrict__ bucket) {
  const int c = blockIdx.x;
  if (c >= nchunks) return;
  const float* __restrict__ src = reinterpret_cast<const float*>(desc[3 * c]);
  float* __restrict__ dst = bucket + desc[3 * c + 1];
  const int n = (int)desc[3 * c + 2];
  if (aligned16(src) && aligned16(dst) && (n & 3) == 0) {
    const float4* s4 = reinterpret_cast<const float4*>(src);
    float4* d4 = reinterpret_cast<float4*>(dst);
    for (int i = threadIdx.x; i < (n >> 2); i += kBlock) {
      float4 a = d4[i];
      const float4 b = s4[i];
      a.x += b.x; a.y += b.y; a.z += b.z; a.w += b.w;
      d4[i] = a;
    }
  } else {
    for (int i = threadIdx.x; i < n; i += kBlock) dst[i] += src[i];
  }
}

__global__ __launch_bounds__(kBlock) void unpack_scale_kernel(
    const int64_t* __restrict__ desc, int nchunks,
    const float* __restrict__ bucket, float scale) {
  const int c = blockIdx.x;
  if (c >= nchunks) return;
  float* __restrict__ dst = reinterpret_cast<float*>(desc[3 * c]);
  const float* __restrict__ src = bucket + desc[3 * c + 1];
  const int n = (int)desc[3 * c + 2];
  for (int i = threadIdx.x; i < n; i += kBlock) dst[i] = src[i] * scale;
}

// ------------------------------------------------------------------- top-k
// |x| threshold count for Gaussian-style top-k selection: count elements with
// |x| >= thr (one pass, device-wide atomic on a counter per threshold).
__global__ __launch_bounds__(kBlock) void count_ge_kernel(
    const float* __restrict__ x, int64_t n, const float* __restrict__ thr,
    int nthr, int32_t* __restrict__ counts) {
  __shared__ int32_t local[32];
  for (int t = threadIdx.x; t < nthr; t += kBlock) local[t] = 0;
  __syncthreads();
  const int64_t stride = (int64_t)gridDim.x * kBlock;
  for (int64_t i = (int64_t)blockIdx.x * kBlock + threadIdx.x; i < n;
       i += stride) {
    float a = fabsf(x[i]);
    for (int t = 0; t < nthr; ++t)
      if (a >= thr[t]) atomicAdd(&local[t], 1);
  }
  __syncthreads();
  for (int t = threadIdx.x; t < nthr; t += kBlock)
    if (local[t]) atomicAdd(&counts[t], local[t]);
}

// Compact indices/values with |x| >= thr into out arrays (cap k).
__global__ __launch_bounds__(kBlock) void select_ge_kernel(
    const float* __restrict__ x, int64_t n, float thr, int64_t cap,
    int64_t* __restrict__ out_idx, float* __restrict__ out_val,
    int32_t* __restrict__ cursor) {
  const int64_t stride = (int64_t)gridDim.x * kBlock;
  for (int64_t i = (int64_t)blockIdx.x * kBlock + threadIdx.x; i < n;
       i += stride) {
    float v = x[i];
    if (fabsf(v) >= thr) {
      int32_t pos = atomicAdd(cursor, 1);
      if (pos < cap) {
        out_idx[pos] = i;
        out_val[pos] = v;
      }
    }
  }
}

int grid_for(int64_t work_items) {
  int64_t blocks = (work_items + kBlock - 1) / kBlock;
  // memory-bound: cap at 8 blocks/CU * 256 CUs, grid-stride the rest
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

}  // namespace

// ------------------------------------------------------------------ bindings
static void fused_sgd(at::Tensor desc, at::Tensor bucket, at::Tensor mom,
                      double lr, double momentum, double dampening, double wd,
                      bool nesterov, double scale, bool first_step,
                      bool maximize) {
  TORCH_CHECK(desc.is_cuda() && desc.dtype() == at::kLong &&
              desc.dim() == 2 && desc.size(1) == 3, "bad desc");
  TORCH_CHECK(bucket.is_cuda() && bucket.dtype() == at::kFloat);
  const int nchunks = (int)desc.size(0);
  const float s = (float)(maximize ? -scale : scale);
  auto stream = c10::hip::getCurrentHIPStream();
  const bool kmom = momentum != 0.0;
#define LAUNCH_SGD(M, N, F)                                                   \
  hipLaunchKernelGGL((fused_sgd_kernel<M, N, F>), dim3(nchunks),              \
                     dim3(kBlock), 0, stream.stream(),                        \
                     desc.data_ptr<int64_t>(), nchunks,                       \
                     bucket.data_ptr<float>(), mom.data_ptr<float>(),         \
                     (float)lr, (float)momentum, (float)dampening, (float)wd, \
                     s)
  if (!kmom) LAUNCH_SGD(false, false, false);
  else if (first_step && nesterov) LAUNCH_SGD(true, true, true);
  else if (first_step) LAUNCH_SGD(true, false, true);
  else if (nesterov) LAUNCH_SGD(true, true, false);
  else LAUNCH_SGD(true, false, false);
#undef LAUNCH_SGD
  HIP_CHECK(hipGetLastError());
}

static void fused_adam(at::Tensor desc, at::Tensor bucket, at::Tensor m,
                       at::Tensor v, double lr, double b1, double b2,
                       double eps, double wd, bool decoupled, double scale,
                       int64_t step) {
  const int nchunks = (int)desc.size(0);
  const double bc1 = 1.0 - std::pow(b1, (double)step);
  const double bc2 = 1.0 - std::pow(b2, (double)step);
  auto stream = c10::hip::getCurrentHIPStream();
  if (decoupled) {
    hipLaunchKernelGGL((fused_adam_kernel<true>), dim3(nchunks), dim3(kBlock),
                       0, stream.stream(), desc.data_ptr<int64_t>(), nchunks,
                       bucket.data_ptr<float>(), m.data_ptr<float>(),
                       v.data_ptr<float>(), (float)lr, (float)b1, (float)b2,
                       (float)eps, (float)wd, (float)scale, (float)(1.0 / bc1),
                       (float)(1.0 / std::sqrt(bc2)));
  } else {
    hipLaunchKernelGGL((fused_adam_kernel<false>), dim3(nchunks), dim3(kBlock),
                       0, stream.stream(), desc.data_ptr<int64_t>(), nchunks,
                       bucket.data_ptr<float>(), m.data_ptr<float>(),
                       v.data_ptr<float>(), (float)lr, (float)b1, (float)b2,
                       (float)eps, (float)wd, (float)scale, (float)(1.0 / bc1),
                       (float)(1.0 / std::sqrt(bc2)));
  }
  HIP_CHECK(hipGetLastError());
}

static void pack(at::Tensor desc, at::Tensor bucket) {
  const int nchunks = (int)desc.size(0);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(pack_kernel, dim3(nchunks), dim3(kBlock), 0,
                     stream.stream(), desc.data_ptr<int64_t>(), nchunks,
                     bucket.data_ptr<float>());
  HIP_CHECK(hipGetLastError());
}

static void pack_add(at::Tensor desc, at::Tensor bucket) {
  const int nchunks = (int)desc.size(0);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(pack_add_kernel, dim3(nchunks), dim3(kBlock), 0,
                     stream.stream(), desc.data_ptr<int64_t>(), nchunks,
                     bucket.data_ptr<float>());
  HIP_CHECK(hipGetLastError());
}

static void unpack_scale(at::Tensor desc, at::Tensor bucket, double scale) {
  const int nchunks = (int)desc.size(0);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(unpack_scale_kernel, dim3(nchunks), dim3(kBlock), 0,
                     stream.stream(), desc.data_ptr<int64_t>(), nchunks,
                     bucket.data_ptr<float>(), (float)scale);
  HIP_CHECK(hipGetLastError());
}

static void count_ge(at::Tensor x, at::Tensor thr, at::Tensor counts) {
  TORCH_CHECK(thr.numel() <= 32, "at most 32 thresholds per pass");
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(count_ge_kernel, dim3(grid_for(x.numel())), dim3(kBlock),
                     0, stream.stream(), x.data_ptr<float>(), x.numel(),
                     thr.data_ptr<float>(), (int)thr.numel(),
                     counts.data_ptr<int32_t>());
  HIP_CHECK(hipGetLastError());
}

static void select_ge(at::Tensor x, double thr, at::Tensor out_idx,
                      at::Tensor out_val, at::Tensor cursor) {
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(select_ge_kernel, dim3(grid_for(x.numel())), dim3(kBlock),
                     0, stream.stream(), x.data_ptr<float>(), x.numel(),
                     (float)thr, out_idx.numel(), out_idx.data_ptr<int64_t>(),
                     out_val.data_ptr<float>(), cursor.data_ptr<int32_t>());
  HIP_CHECK(hipGetLastError());
}

// ---------------------------------------------------------- fused BN (NHWC)
extern "C" int dear_bn_nparts(long, int);
extern "C" void dear_bn_fwd(hipStream_t, const float*, const float*, float*,
                            const float*, const float*, float*, float*,
                            float*, float*, float*, float*, int, long, int,
                            float, float, int, int);
extern "C" void dear_bn_bwd(hipStream_t, const float*, const float*,
                            const float*, float*, const float*, const float*,
                            const float*, const float*, float*, float*, int,
                            float*, float*, float*, long, int, int, int);

static void bn_fwd(at::Tensor x, c10::optional<at::Tensor> res, at::Tensor y,
                   at::Tensor w, at::Tensor b, at::Tensor mean,
                   at::Tensor invstd, at::Tensor rmean, at::Tensor rvar,
                   at::Tensor psum, at::Tensor psumsq, int64_t rows, int64_t C,
                   double eps, double momentum, bool training, bool relu) {
  auto stream = c10::hip::getCurrentHIPStream();
  dear_bn_fwd(stream.stream(), x.data_ptr<float>(),
              res ? res->data_ptr<float>() : nullptr, y.data_ptr<float>(),
              w.data_ptr<float>(), b.data_ptr<float>(),
              mean.data_ptr<float>(), invstd.data_ptr<float>(),
              rmean.data_ptr<float>(), rvar.data_ptr<float>(),
              psum.data_ptr<float>(), psumsq.data_ptr<float>(),
              (int)psum.size(0), (long)rows, (int)C, (float)eps,
              (float)momentum, training ? 1 : 0, relu ? 1 : 0);
  HIP_CHECK(hipGetLastError());
}

static void bn_bwd(at::Tensor x, at::Tensor dy, at::Tensor y,
                   c10::optional<at::Tensor> dy_eff, at::Tensor w,
                   at::Tensor b, at::Tensor mean, at::Tensor invstd,
                   at::Tensor pdb, at::Tensor pdg, at::Tensor dbeta,
                   at::Tensor dgamma, at::Tensor dx, int64_t rows, int64_t C,
                   bool relu, bool want_dy_eff) {
  auto stream = c10::hip::getCurrentHIPStream();
  dear_bn_bwd(stream.stream(), x.data_ptr<float>(), dy.data_ptr<float>(),
              y.data_ptr<float>(),
              dy_eff ? dy_eff->data_ptr<float>() : nullptr,
              w.data_ptr<float>(), b.data_ptr<float>(),
              mean.data_ptr<float>(),
              invstd.data_ptr<float>(), pdb.data_ptr<float>(),
              pdg.data_ptr<float>(), (int)pdb.size(0),
              dbeta.data_ptr<float>(), dgamma.data_ptr<float>(),
              dx.data_ptr<float>(), (long)rows, (int)C, relu ? 1 : 0,
              want_dy_eff ? 1 : 0);
  HIP_CHECK(hipGetLastError());
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "DeAR CDNA4 fused kernels (gfx950)";
  m.def("bn_nparts", [](int64_t rows, int64_t C) {
    return dear_bn_nparts((long)rows, (int)C);
  });
  m.def("bn_fwd", &bn_fwd);
  m.def("bn_bwd", &bn_bwd);
  m.def("fused_sgd", &fused_sgd);
  m.def("fused_adam", &fused_adam);
  m.def("pack", &pack);
  m.def("pack_add", &pack_add);
  m.def("unpack_scale", &unpack_scale);
  m.def("count_ge", &count_ge);
  m.def("select_ge", &select_ge);
}
