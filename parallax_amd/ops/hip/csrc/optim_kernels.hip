// Fused multi-tensor optimizers + sparse-row optimizers + grad-clip/EMA.
//
// The reference's update-op vocabulary (graph_transform_lib.py:56-75):
// dense ApplyGradientDescent/Momentum/Adagrad/Adam/RMSProp as ONE
// multi-tensor launch per optimizer step (up to MT_MAX tensors per launch,
// grid-stride over the summed element count, binary search over the
// cumulative-offset table in kernel args to locate the tensor), and
// SparseApplyAdagrad / sparse SGD as per-row kernels over deduplicated
// (ids, grad rows).
//
// Param/grad dtype: fp32 or bf16 (fp32 math); optimizer state always fp32.

#include <torch/extension.h>
#include <ATen/ATen.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

#define MT_MAX 32

struct MTMeta {
  void* a[MT_MAX];    // param
  void* b[MT_MAX];    // grad
  void* c[MT_MAX];    // state 1 (momentum / accum / m / square_avg)
  void* d[MT_MAX];    // state 2 (v / momentum buf)
  float e[MT_MAX];    // per-tensor scalar (adam step count)
  long cum[MT_MAX + 1];
  int n;
};

__device__ __forceinline__ int mt_find(const MTMeta& m, long idx) {
  int lo = 0, hi = m.n - 1;
  while (lo < hi) {
    int mid = (lo + hi) >> 1;
    if (idx < m.cum[mid + 1]) hi = mid; else lo = mid + 1;
  }
  return lo;
}

// Functor-driven elementwise multi-tensor kernel.  OP provides
// apply(ti, j, meta, args...).
template <typename T, typename OP>
__global__ void mt_kernel(MTMeta meta, OP op) {
  const long total = meta.cum[meta.n];
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const int ti = mt_find(meta, idx);
    const long j = idx - meta.cum[ti];
    op.template apply<T>(meta, ti, j);
  }
}

// --- op functors -----------------------------------------------------------

struct SGDOp {
  float lr, wd;
  template <typename T>
  __device__ void apply(const MTMeta& m, int ti, long j) const {
    T* p = reinterpret_cast<T*>(m.a[ti]);
    const T* g = reinterpret_cast<const T*>(m.b[ti]);
    float gv = to_f32<T>(g[j]) + wd * to_f32<T>(p[j]);
    p[j] = from_f32<T>(to_f32<T>(p[j]) - lr * gv);
  }
};

struct MomentumOp {
  float lr, mom, wd;
  bool nesterov;
  template <typename T>
  __device__ void apply(const MTMeta& m, int ti, long j) const {
    T* p = reinterpret_cast<T*>(m.a[ti]);
    const T* g = reinterpret_cast<const T*>(m.b[ti]);
    float* buf = reinterpret_cast<float*>(m.c[ti]);
    float gv = to_f32<T>(g[j]) + wd * to_f32<T>(p[j]);
    float b = buf[j] * mom + gv;
    buf[j] = b;
    float upd = nesterov ? (gv + mom * b) : b;
    p[j] = from_f32<T>(to_f32<T>(p[j]) - lr * upd);
  }
};

struct AdagradOp {
  float lr, eps;
  template <typename T>
  __device__ void apply(const MTMeta& m, int ti, long j) const {
    T* p = reinterpret_cast<T*>(m.a[ti]);
    const T* g = reinterpret_cast<const T*>(m.b[ti]);
    float* acc = reinterpret_cast<float*>(m.c[ti]);
    float gv = to_f32<T>(g[j]);
    float a = acc[j] + gv * gv;
    acc[j] = a;
    p[j] = from_f32<T>(to_f32<T>(p[j]) - lr * gv / (sqrtf(a) + eps));
  }
};

struct AdamOp {
  float lr, b1, b2, eps, wd;
  const float* t_dev;  // device step counter (hipGraph-capturable); when
                       // null the host scalar e[ti] is used instead
  template <typename T>
  __device__ void apply(const MTMeta& m, int ti, long j) const {
    T* p = reinterpret_cast<T*>(m.a[ti]);
    const T* g = reinterpret_cast<const T*>(m.b[ti]);
    float* mv = reinterpret_cast<float*>(m.c[ti]);
    float* vv = reinterpret_cast<float*>(m.d[ti]);
    const float t = t_dev ? *t_dev : m.e[ti];
    float gv = to_f32<T>(g[j]) + wd * to_f32<T>(p[j]);
    float m1 = mv[j] * b1 + (1.f - b1) * gv;
    float v1 = vv[j] * b2 + (1.f - b2) * gv * gv;
    mv[j] = m1;
    vv[j] = v1;
    const float bc1 = 1.f - powf(b1, t);
    const float bc2 = 1.f - powf(b2, t);
    float denom = sqrtf(v1 / bc2) + eps;
    p[j] = from_f32<T>(to_f32<T>(p[j]) - (lr / bc1) * m1 / denom);
  }
};

struct RMSPropOp {
  float lr, decay, mom, eps;
  template <typename T>
  __device__ void apply(const MTMeta& m, int ti, long j) const {
    T* p = reinterpret_cast<T*>(m.a[ti]);
    const T* g = reinterpret_cast<const T*>(m.b[ti]);
    float* sq = reinterpret_cast<float*>(m.c[ti]);
    float* buf = reinterpret_cast<float*>(m.d[ti]);
    float gv = to_f32<T>(g[j]);
    float s = sq[j] * decay + (1.f - decay) * gv * gv;
    sq[j] = s;
    float upd = gv / (sqrtf(s) + eps);
    float b = buf[j] * mom + upd;
    buf[j] = b;
    p[j] = from_f32<T>(to_f32<T>(p[j]) - lr * b);
  }
};

struct EMAOp {
  float decay;
  template <typename T>
  __device__ void apply(const MTMeta& m, int ti, long j) const {
    T* shadow = reinterpret_cast<T*>(m.a[ti]);
    const T* val = reinterpret_cast<const T*>(m.b[ti]);
    shadow[j] = from_f32<T>(decay * to_f32<T>(shadow[j]) +
                            (1.f - decay) * to_f32<T>(val[j]));
  }
};

struct ScaleOp {
  const float* norm;  // device scalar
  float clip;
  template <typename T>
  __device__ void apply(const MTMeta& m, int ti, long j) const {
    T* t = reinterpret_cast<T*>(m.a[ti]);
    float n = *norm;
    float scale = clip / fmaxf(n, clip);
    t[j] = from_f32<T>(to_f32<T>(t[j]) * scale);
  }
};

// --- host-side launch plumbing ---------------------------------------------

template <typename OP>
static void mt_launch(const std::vector<at::Tensor>& a,
                      const std::vector<at::Tensor>& b,
                      const std::vector<at::Tensor>& c,
                      const std::vector<at::Tensor>& d,
                      const std::vector<float>& e, OP op) {
  TORCH_CHECK(!a.empty());
  auto stream = at::cuda::getCurrentCUDAStream();
  size_t i = 0;
  while (i < a.size()) {
    MTMeta meta;
    meta.n = 0;
    meta.cum[0] = 0;
    while (i < a.size() && meta.n < MT_MAX) {
      int k = meta.n;
      meta.a[k] = a[i].data_ptr();
      meta.b[k] = b.empty() ? nullptr : b[i].data_ptr();
      meta.c[k] = c.empty() ? nullptr : c[i].data_ptr();
      meta.d[k] = d.empty() ? nullptr : d[i].data_ptr();
      meta.e[k] = e.empty() ? 0.f : e[i];
      meta.cum[k + 1] = meta.cum[k] + a[i].numel();
      meta.n++;
      i++;
    }
    const long total = meta.cum[meta.n];
    AT_DISPATCH_FLOATING_TYPES_AND2(
        at::kBFloat16, at::kHalf, a[0].scalar_type(), "mt_launch", [&] {
          mt_kernel<scalar_t, OP>
              <<<px_grid(total), PX_BLOCK, 0, stream>>>(meta, op);
          HIP_CHECK_KERNEL();
        });
  }
}

// The mt_* kernels sweep each tensor's STORAGE linearly, which is correct
// for any non-overlapping dense layout (e.g. channels_last conv weights)
// as long as every companion tensor (grad, state) shares the param's
// strides — autograd and zeros_like both preserve them.
static void check_contig(const std::vector<at::Tensor>& ts,
                         const char* name) {
  for (auto& t : ts) {
    TORCH_CHECK(t.is_non_overlapping_and_dense(), name,
                ": tensors must be non-overlapping and dense");
  }
}

// strides of size-1 dims are meaningless (a [cout,cin,1,1] conv weight in
// channels_last and its contiguous grad share the same storage order) —
// compare layouts over dims with extent > 1 only.
static bool same_dense_layout(const at::Tensor& x, const at::Tensor& y) {
  if (x.sizes() != y.sizes()) return false;
  for (int64_t i = 0; i < x.dim(); ++i) {
    if (x.size(i) > 1 && x.stride(i) != y.stride(i)) return false;
  }
  return true;
}

static void check_same_layout(const std::vector<at::Tensor>& a,
                              const std::vector<at::Tensor>& b,
                              const char* name) {
  TORCH_CHECK(a.size() == b.size(), name, ": list size mismatch");
  for (size_t i = 0; i < a.size(); ++i) {
    TORCH_CHECK(same_dense_layout(a[i], b[i]),
                name, ": layout mismatch at tensor ", i);
  }
}

void multi_tensor_sgd(std::vector<at::Tensor> params,
                      std::vector<at::Tensor> grads, double lr, double wd) {
  check_contig(params, "mt_sgd");
  check_same_layout(params, grads, "mt_sgd");
  mt_launch(params, grads, {}, {}, {}, SGDOp{(float)lr, (float)wd});
}

void multi_tensor_momentum(std::vector<at::Tensor> params,
                           std::vector<at::Tensor> grads,
                           std::vector<at::Tensor> bufs, double lr,
                           double mom, double wd, bool nesterov) {
  check_contig(params, "mt_momentum");
  check_same_layout(params, grads, "mt_momentum");
  check_same_layout(params, bufs, "mt_momentum");
  mt_launch(params, grads, bufs, {}, {},
            MomentumOp{(float)lr, (float)mom, (float)wd, nesterov});
}

void multi_tensor_adagrad(std::vector<at::Tensor> params,
                          std::vector<at::Tensor> grads,
                          std::vector<at::Tensor> accums, double lr,
                          double eps) {
  check_contig(params, "mt_adagrad");
  check_same_layout(params, grads, "mt_adagrad");
  mt_launch(params, grads, accums, {}, {},
            AdagradOp{(float)lr, (float)eps});
}

void multi_tensor_adam(std::vector<at::Tensor> params,
                       std::vector<at::Tensor> grads,
                       std::vector<at::Tensor> ms,
                       std::vector<at::Tensor> vs, std::vector<double> steps,
                       double lr, double b1, double b2, double eps,
                       double wd) {
  check_contig(params, "mt_adam");
  check_same_layout(params, grads, "mt_adam");
  std::vector<float> e(steps.begin(), steps.end());
  mt_launch(params, grads, ms, vs, e,
            AdamOp{(float)lr, (float)b1, (float)b2, (float)eps, (float)wd,
                   nullptr});
}

// hipGraph-capturable Adam: the step count lives in a 0-dim fp32 DEVICE
// tensor the kernel dereferences at run time, so a captured optimizer
// graph replays with fresh bias correction every step (the host-scalar
// variant above bakes t into the kernel args at capture time).
void multi_tensor_adam_dev(std::vector<at::Tensor> params,
                           std::vector<at::Tensor> grads,
                           std::vector<at::Tensor> ms,
                           std::vector<at::Tensor> vs, at::Tensor step_dev,
                           double lr, double b1, double b2, double eps,
                           double wd) {
  check_contig(params, "mt_adam_dev");
  check_same_layout(params, grads, "mt_adam_dev");
  TORCH_CHECK(step_dev.is_cuda() &&
                  step_dev.scalar_type() == at::kFloat &&
                  step_dev.numel() == 1,
              "mt_adam_dev: step_dev must be a 0-dim fp32 CUDA tensor");
  mt_launch(params, grads, ms, vs, {},
            AdamOp{(float)lr, (float)b1, (float)b2, (float)eps, (float)wd,
                   step_dev.data_ptr<float>()});
}

void multi_tensor_rmsprop(std::vector<at::Tensor> params,
                          std::vector<at::Tensor> grads,
                          std::vector<at::Tensor> sqs,
                          std::vector<at::Tensor> bufs, double lr,
                          double decay, double mom, double eps) {
  check_contig(params, "mt_rmsprop");
  check_same_layout(params, grads, "mt_rmsprop");
  mt_launch(params, grads, sqs, bufs, {},
            RMSPropOp{(float)lr, (float)decay, (float)mom, (float)eps});
}

void multi_tensor_ema(std::vector<at::Tensor> shadows,
                      std::vector<at::Tensor> values, double decay) {
  check_contig(shadows, "mt_ema");
  mt_launch(shadows, values, {}, {}, {}, EMAOp{(float)decay});
}

// --- global norm -----------------------------------------------------------

template <typename T>
__global__ void mt_sqsum_kernel(MTMeta meta, float* out) {
  __shared__ float scratch[PX_BLOCK / WAVE];
  const long total = meta.cum[meta.n];
  float acc = 0.f;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const int ti = mt_find(meta, idx);
    const long j = idx - meta.cum[ti];
    float v = to_f32<T>(reinterpret_cast<const T*>(meta.a[ti])[j]);
    acc += v * v;
  }
  acc = block_reduce_sum(acc, scratch);
  if (threadIdx.x == 0) atomicAdd(out, acc);
}

at::Tensor multi_tensor_l2norm(std::vector<at::Tensor> tensors) {
  TORCH_CHECK(!tensors.empty());
  check_contig(tensors, "mt_l2norm");
  auto out = at::zeros({}, tensors[0].options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  size_t i = 0;
  while (i < tensors.size()) {
    MTMeta meta;
    meta.n = 0;
    meta.cum[0] = 0;
    while (i < tensors.size() && meta.n < MT_MAX) {
      int k = meta.n;
      meta.a[k] = tensors[i].data_ptr();
      meta.cum[k + 1] = meta.cum[k] + tensors[i].numel();
      meta.n++;
      i++;
    }
    AT_DISPATCH_FLOATING_TYPES_AND2(
        at::kBFloat16, at::kHalf, tensors[0].scalar_type(), "mt_l2norm",
        [&] {
          mt_sqsum_kernel<scalar_t>
              <<<px_grid(meta.cum[meta.n]), PX_BLOCK, 0, stream>>>(
                  meta, out.data_ptr<float>());
          HIP_CHECK_KERNEL();
        });
  }
  return out.sqrt();
}

void multi_tensor_scale_by_norm(std::vector<at::Tensor> tensors,
                                at::Tensor norm, double clip) {
  check_contig(tensors, "mt_scale");
  auto n32 = norm.to(at::kFloat).contiguous();
  mt_launch(tensors, {}, {}, {}, {},
            ScaleOp{n32.data_ptr<float>(), (float)clip});
}

// --- sparse row optimizers (SparseApplyAdagrad etc.) ------------------------

template <typename T>
__global__ void sparse_adagrad_kernel(T* __restrict__ p,
                                      float* __restrict__ acc,
                                      const int64_t* __restrict__ ids,
                                      const T* __restrict__ grows, long u,
                                      long d, float lr, float eps) {
  const long total = u * d;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long r = idx / d;
    const long c = idx - r * d;
    const long row = ids[r];
    float gv = to_f32<T>(grows[r * d + c]);
    float a = acc[row * d + c] + gv * gv;
    acc[row * d + c] = a;
    p[row * d + c] =
        from_f32<T>(to_f32<T>(p[row * d + c]) - lr * gv / (sqrtf(a) + eps));
  }
}

void sparse_adagrad(at::Tensor param, at::Tensor accum, at::Tensor ids,
                    at::Tensor grad_rows, double lr, double eps) {
  TORCH_CHECK(param.is_cuda() && param.is_contiguous());
  TORCH_CHECK(accum.scalar_type() == at::kFloat);
  const long u = ids.numel();
  if (u == 0) return;
  const long d = param.dim() == 1 ? 1 : param.size(1);
  auto g = grad_rows.contiguous();
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, param.scalar_type(), "sparse_adagrad", [&] {
        sparse_adagrad_kernel<scalar_t>
            <<<px_grid(u * d), PX_BLOCK, 0, stream>>>(
                reinterpret_cast<scalar_t*>(param.data_ptr()),
                accum.data_ptr<float>(), ids.data_ptr<int64_t>(),
                reinterpret_cast<const scalar_t*>(g.data_ptr()), u, d,
                (float)lr, (float)eps);
        HIP_CHECK_KERNEL();
      });
}

template <typename T>
__global__ void sparse_sgd_kernel(T* __restrict__ p,
                                  const int64_t* __restrict__ ids,
                                  const T* __restrict__ grows, long u, long d,
                                  float lr) {
  const long total = u * d;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long r = idx / d;
    const long c = idx - r * d;
    const long row = ids[r];
    p[row * d + c] = from_f32<T>(to_f32<T>(p[row * d + c]) -
                                 lr * to_f32<T>(grows[r * d + c]));
  }
}

void sparse_sgd(at::Tensor param, at::Tensor ids, at::Tensor grad_rows,
                double lr) {
  TORCH_CHECK(param.is_cuda() && param.is_contiguous());
  const long u = ids.numel();
  if (u == 0) return;
  const long d = param.dim() == 1 ? 1 : param.size(1);
  auto g = grad_rows.contiguous();
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, param.scalar_type(), "sparse_sgd", [&] {
        sparse_sgd_kernel<scalar_t>
            <<<px_grid(u * d), PX_BLOCK, 0, stream>>>(
                reinterpret_cast<scalar_t*>(param.data_ptr()),
                ids.data_ptr<int64_t>(),
                reinterpret_cast<const scalar_t*>(g.data_ptr()), u, d,
                (float)lr);
        HIP_CHECK_KERNEL();
      });
}

// --- fusion-bucket pack/unpack ---------------------------------------------
// One launch packs a bucket's per-param gradients into the flat bf16/fp32
// fusion buffer (and one unpacks the reduced buffer back, scaled by
// 1/world) — replaces per-parameter copy/div kernel pairs in the dense
// all-reduce engine (2 x n_params launches -> 2 per <=32-tensor chunk).

template <typename TSrc, typename TDst>
__global__ void mt_pack_kernel(MTMeta meta, TDst* __restrict__ flat) {
  const long total = meta.cum[meta.n];
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const int ti = mt_find(meta, idx);
    const long j = idx - meta.cum[ti];
    flat[idx] = from_f32<TDst>(
        to_f32<TSrc>(reinterpret_cast<const TSrc*>(meta.a[ti])[j]));
  }
}

template <typename TSrc, typename TDst>
__global__ void mt_unpack_kernel(MTMeta meta, const TSrc* __restrict__ flat,
                                 float scale) {
  const long total = meta.cum[meta.n];
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const int ti = mt_find(meta, idx);
    const long j = idx - meta.cum[ti];
    reinterpret_cast<TDst*>(meta.a[ti])[j] =
        from_f32<TDst>(to_f32<TSrc>(flat[idx]) * scale);
  }
}

template <bool PACK>
static void mt_pack_unpack(std::vector<at::Tensor>& tensors,
                           at::Tensor& flat, double scale) {
  check_contig(tensors, "mt_pack");
  auto stream = at::cuda::getCurrentCUDAStream();
  size_t i = 0;
  long base = 0;
  while (i < tensors.size()) {
    MTMeta meta;
    meta.n = 0;
    meta.cum[0] = 0;
    while (i < tensors.size() && meta.n < MT_MAX) {
      int k = meta.n;
      meta.a[k] = tensors[i].data_ptr();
      meta.cum[k + 1] = meta.cum[k] + tensors[i].numel();
      meta.n++;
      i++;
    }
    const long total = meta.cum[meta.n];
    AT_DISPATCH_FLOATING_TYPES_AND2(
        at::kBFloat16, at::kHalf, tensors[0].scalar_type(), "mt_pack_t",
        [&] {
          using TP = scalar_t;
          AT_DISPATCH_FLOATING_TYPES_AND2(
              at::kBFloat16, at::kHalf, flat.scalar_type(), "mt_pack_f",
              [&] {
                using TF = scalar_t;
                if (PACK) {
                  mt_pack_kernel<TP, TF>
                      <<<px_grid(total), PX_BLOCK, 0, stream>>>(
                          meta,
                          reinterpret_cast<TF*>(flat.data_ptr()) + base);
                } else {
                  mt_unpack_kernel<TF, TP>
                      <<<px_grid(total), PX_BLOCK, 0, stream>>>(
                          meta,
                          reinterpret_cast<const TF*>(flat.data_ptr())
                              + base,
                          (float)scale);
                }
                HIP_CHECK_KERNEL();
              });
        });
    base += total;
  }
}

void multi_tensor_pack(std::vector<at::Tensor> tensors, at::Tensor flat) {
  mt_pack_unpack<true>(tensors, flat, 1.0);
}

void multi_tensor_unpack(std::vector<at::Tensor> tensors, at::Tensor flat,
                         double scale) {
  mt_pack_unpack<false>(tensors, flat, scale);
}
