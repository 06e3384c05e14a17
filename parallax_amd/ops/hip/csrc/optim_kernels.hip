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
  void* f[MT_MAX];    // fp32 master weight (nullptr when param IS fp32):
                      // the update reads/writes the master and stores the
                      // rounded copy into the (bf16) param — pure-bf16
                      // training without per-step cast kernels
  float e[MT_MAX];    // per-tensor scalar (adam step count)
  long cum[MT_MAX + 1];
  long vcum[MT_MAX + 1];  // cumulative count of 4-element vector chunks
  int n;
};

__device__ __forceinline__ int mt_find_in(const long* cum, int n, long idx) {
  int lo = 0, hi = n - 1;
  while (lo < hi) {
    int mid = (lo + hi) >> 1;
    if (idx < cum[mid + 1]) hi = mid; else lo = mid + 1;
  }
  return lo;
}

__device__ __forceinline__ int mt_find(const MTMeta& m, long idx) {
  return mt_find_in(m.cum, m.n, idx);
}

// 4-wide vector of T with natural vector alignment (float4 -> dwordx4
// loads, bf16x4 -> dwordx2) for the vectorized multi-tensor path.
template <typename T>
struct alignas(4 * sizeof(T)) V4 {
  T v[4];
};

template <typename T>
__device__ __forceinline__ V4<T> ld4(const T* p, long j) {
  return *reinterpret_cast<const V4<T>*>(p + j);
}

template <typename T>
__device__ __forceinline__ void st4(T* p, long j, const V4<T>& x) {
  *reinterpret_cast<V4<T>*>(p + j) = x;
}

// Param read/write through the optional fp32 master (meta.f[ti]).
template <typename T>
__device__ __forceinline__ float ldp(const MTMeta& m, int ti, long j,
                                     const T* p) {
  const float* mw = reinterpret_cast<const float*>(m.f[ti]);
  return mw ? mw[j] : to_f32<T>(p[j]);
}

template <typename T>
__device__ __forceinline__ void stp(const MTMeta& m, int ti, long j, T* p,
                                    float r) {
  float* mw = reinterpret_cast<float*>(m.f[ti]);
  if (mw) mw[j] = r;
  p[j] = from_f32<T>(r);
}

template <typename T>
__device__ __forceinline__ void ldp4(const MTMeta& m, int ti, long j,
                                     const T* p, float out[4]) {
  const float* mw = reinterpret_cast<const float*>(m.f[ti]);
  if (mw) {
    V4<float> v = ld4(mw, j);
#pragma unroll
    for (int k = 0; k < 4; ++k) out[k] = v.v[k];
  } else {
    V4<T> v = ld4(p, j);
#pragma unroll
    for (int k = 0; k < 4; ++k) out[k] = to_f32<T>(v.v[k]);
  }
}

template <typename T>
__device__ __forceinline__ void stp4(const MTMeta& m, int ti, long j, T* p,
                                     const float r[4]) {
  float* mw = reinterpret_cast<float*>(m.f[ti]);
  if (mw) {
    V4<float> v;
#pragma unroll
    for (int k = 0; k < 4; ++k) v.v[k] = r[k];
    st4(mw, j, v);
  }
  V4<T> v;
#pragma unroll
  for (int k = 0; k < 4; ++k) v.v[k] = from_f32<T>(r[k]);
  st4(p, j, v);
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

// Vectorized variant: grid-stride over 4-element chunks (vcum index
// space, one binary search per chunk, dwordx4/x2 accesses); the <=3
// tail elements of each tensor are finished by block 0's first lanes.
// Requires every tensor pointer 16-byte aligned (checked on host).
template <typename T, typename OP>
__global__ void mt_kernel_v4(MTMeta meta, OP op) {
  const long vtotal = meta.vcum[meta.n];
  for (long vidx = (long)blockIdx.x * blockDim.x + threadIdx.x;
       vidx < vtotal; vidx += (long)gridDim.x * blockDim.x) {
    const int ti = mt_find_in(meta.vcum, meta.n, vidx);
    const long j = 4 * (vidx - meta.vcum[ti]);
    op.template apply4<T>(meta, ti, j);
  }
  if (blockIdx.x == 0 && threadIdx.x < 4 * (unsigned)meta.n) {
    const int ti = threadIdx.x >> 2;
    const long numel = meta.cum[ti + 1] - meta.cum[ti];
    const long j = (numel & ~3L) + (threadIdx.x & 3);
    if (j < numel) op.template apply<T>(meta, ti, j);
  }
}

// --- op functors -----------------------------------------------------------
//
// Each op provides the scalar per-element math (elem), the scalar global
// apply (tails + misaligned fallback), and apply4 — explicit V4 loads /
// stores (dwordx4 for fp32, dwordx2 for bf16) with the math unrolled over
// the four lanes.  `vec` gates the vectorized kernel.

struct SGDOp {
  static constexpr bool vec = true;
  float lr, wd;
  __device__ __forceinline__ float elem(float pv, float gv) const {
    return pv - lr * (gv + wd * pv);
  }
  template <typename T>
  __device__ void apply(const MTMeta& m, int ti, long j) const {
    T* p = reinterpret_cast<T*>(m.a[ti]);
    const T* g = reinterpret_cast<const T*>(m.b[ti]);
    stp(m, ti, j, p, elem(ldp(m, ti, j, p), to_f32<T>(g[j])));
  }
  template <typename T>
  __device__ void apply4(const MTMeta& m, int ti, long j) const {
    T* p = reinterpret_cast<T*>(m.a[ti]);
    const T* g = reinterpret_cast<const T*>(m.b[ti]);
    float pw[4];
    ldp4(m, ti, j, p, pw);
    V4<T> gv = ld4(g, j);
#pragma unroll
    for (int k = 0; k < 4; ++k) pw[k] = elem(pw[k], to_f32<T>(gv.v[k]));
    stp4(m, ti, j, p, pw);
  }
};

struct MomentumOp {
  static constexpr bool vec = true;
  float lr, mom, wd;
  bool nesterov;
  __device__ __forceinline__ float elem(float pv, float gv,
                                        float& buf) const {
    float gw = gv + wd * pv;
    float b = buf * mom + gw;
    buf = b;
    float upd = nesterov ? (gw + mom * b) : b;
    return pv - lr * upd;
  }
  template <typename T>
  __device__ void apply(const MTMeta& m, int ti, long j) const {
    T* p = reinterpret_cast<T*>(m.a[ti]);
    const T* g = reinterpret_cast<const T*>(m.b[ti]);
    float* buf = reinterpret_cast<float*>(m.c[ti]);
    stp(m, ti, j, p, elem(ldp(m, ti, j, p), to_f32<T>(g[j]), buf[j]));
  }
  template <typename T>
  __device__ void apply4(const MTMeta& m, int ti, long j) const {
    T* p = reinterpret_cast<T*>(m.a[ti]);
    const T* g = reinterpret_cast<const T*>(m.b[ti]);
    float* buf = reinterpret_cast<float*>(m.c[ti]);
    float pw[4];
    ldp4(m, ti, j, p, pw);
    V4<T> gv = ld4(g, j);
    V4<float> bv = ld4(buf, j);
#pragma unroll
    for (int k = 0; k < 4; ++k)
      pw[k] = elem(pw[k], to_f32<T>(gv.v[k]), bv.v[k]);
    stp4(m, ti, j, p, pw);
    st4(buf, j, bv);
  }
};

struct AdagradOp {
  static constexpr bool vec = true;
  float lr, eps;
  __device__ __forceinline__ float elem(float pv, float gv,
                                        float& acc) const {
    float a = acc + gv * gv;
    acc = a;
    return pv - lr * gv / (sqrtf(a) + eps);
  }
  template <typename T>
  __device__ void apply(const MTMeta& m, int ti, long j) const {
    T* p = reinterpret_cast<T*>(m.a[ti]);
    const T* g = reinterpret_cast<const T*>(m.b[ti]);
    float* acc = reinterpret_cast<float*>(m.c[ti]);
    stp(m, ti, j, p, elem(ldp(m, ti, j, p), to_f32<T>(g[j]), acc[j]));
  }
  template <typename T>
  __device__ void apply4(const MTMeta& m, int ti, long j) const {
    T* p = reinterpret_cast<T*>(m.a[ti]);
    const T* g = reinterpret_cast<const T*>(m.b[ti]);
    float* acc = reinterpret_cast<float*>(m.c[ti]);
    float pw[4];
    ldp4(m, ti, j, p, pw);
    V4<T> gv = ld4(g, j);
    V4<float> av = ld4(acc, j);
#pragma unroll
    for (int k = 0; k < 4; ++k)
      pw[k] = elem(pw[k], to_f32<T>(gv.v[k]), av.v[k]);
    stp4(m, ti, j, p, pw);
    st4(acc, j, av);
  }
};

struct AdamOp {
  static constexpr bool vec = true;
  float lr, b1, b2, eps, wd;
  const float* t_dev;  // device step counter (hipGraph-capturable); when
                       // null the host scalar e[ti] is used instead
  __device__ __forceinline__ float elem(float pv, float gv, float& mv,
                                        float& vv, float bc1,
                                        float bc2) const {
    float gw = gv + wd * pv;
    float m1 = mv * b1 + (1.f - b1) * gw;
    float v1 = vv * b2 + (1.f - b2) * gw * gw;
    mv = m1;
    vv = v1;
    float denom = sqrtf(v1 / bc2) + eps;
    return pv - (lr / bc1) * m1 / denom;
  }
  template <typename T>
  __device__ void apply(const MTMeta& m, int ti, long j) const {
    T* p = reinterpret_cast<T*>(m.a[ti]);
    const T* g = reinterpret_cast<const T*>(m.b[ti]);
    float* mv = reinterpret_cast<float*>(m.c[ti]);
    float* vv = reinterpret_cast<float*>(m.d[ti]);
    const float t = t_dev ? *t_dev : m.e[ti];
    const float bc1 = 1.f - powf(b1, t);
    const float bc2 = 1.f - powf(b2, t);
    stp(m, ti, j, p,
        elem(ldp(m, ti, j, p), to_f32<T>(g[j]), mv[j], vv[j], bc1, bc2));
  }
  template <typename T>
  __device__ void apply4(const MTMeta& m, int ti, long j) const {
    T* p = reinterpret_cast<T*>(m.a[ti]);
    const T* g = reinterpret_cast<const T*>(m.b[ti]);
    float* mp = reinterpret_cast<float*>(m.c[ti]);
    float* vp = reinterpret_cast<float*>(m.d[ti]);
    const float t = t_dev ? *t_dev : m.e[ti];
    const float bc1 = 1.f - powf(b1, t);
    const float bc2 = 1.f - powf(b2, t);
    float pw[4];
    ldp4(m, ti, j, p, pw);
    V4<T> gv = ld4(g, j);
    V4<float> mv = ld4(mp, j), vv = ld4(vp, j);
#pragma unroll
    for (int k = 0; k < 4; ++k)
      pw[k] = elem(pw[k], to_f32<T>(gv.v[k]), mv.v[k], vv.v[k], bc1, bc2);
    stp4(m, ti, j, p, pw);
    st4(mp, j, mv);
    st4(vp, j, vv);
  }
};

struct RMSPropOp {
  static constexpr bool vec = true;
  float lr, decay, mom, eps;
  __device__ __forceinline__ float elem(float pv, float gv, float& sq,
                                        float& buf) const {
    float s = sq * decay + (1.f - decay) * gv * gv;
    sq = s;
    float upd = gv / (sqrtf(s) + eps);
    float b = buf * mom + upd;
    buf = b;
    return pv - lr * b;
  }
  template <typename T>
  __device__ void apply(const MTMeta& m, int ti, long j) const {
    T* p = reinterpret_cast<T*>(m.a[ti]);
    const T* g = reinterpret_cast<const T*>(m.b[ti]);
    float* sq = reinterpret_cast<float*>(m.c[ti]);
    float* buf = reinterpret_cast<float*>(m.d[ti]);
    stp(m, ti, j, p,
        elem(ldp(m, ti, j, p), to_f32<T>(g[j]), sq[j], buf[j]));
  }
  template <typename T>
  __device__ void apply4(const MTMeta& m, int ti, long j) const {
    T* p = reinterpret_cast<T*>(m.a[ti]);
    const T* g = reinterpret_cast<const T*>(m.b[ti]);
    float* sp = reinterpret_cast<float*>(m.c[ti]);
    float* bp = reinterpret_cast<float*>(m.d[ti]);
    float pw[4];
    ldp4(m, ti, j, p, pw);
    V4<T> gv = ld4(g, j);
    V4<float> sv = ld4(sp, j), bv = ld4(bp, j);
#pragma unroll
    for (int k = 0; k < 4; ++k)
      pw[k] = elem(pw[k], to_f32<T>(gv.v[k]), sv.v[k], bv.v[k]);
    stp4(m, ti, j, p, pw);
    st4(sp, j, sv);
    st4(bp, j, bv);
  }
};

struct EMAOp {
  static constexpr bool vec = true;
  float decay;
  __device__ __forceinline__ float elem(float sv, float vv) const {
    return decay * sv + (1.f - decay) * vv;
  }
  template <typename T>
  __device__ void apply(const MTMeta& m, int ti, long j) const {
    T* shadow = reinterpret_cast<T*>(m.a[ti]);
    const T* val = reinterpret_cast<const T*>(m.b[ti]);
    shadow[j] = from_f32<T>(elem(to_f32<T>(shadow[j]), to_f32<T>(val[j])));
  }
  template <typename T>
  __device__ void apply4(const MTMeta& m, int ti, long j) const {
    T* shadow = reinterpret_cast<T*>(m.a[ti]);
    const T* val = reinterpret_cast<const T*>(m.b[ti]);
    V4<T> sv = ld4(shadow, j), vv = ld4(val, j);
#pragma unroll
    for (int k = 0; k < 4; ++k)
      sv.v[k] = from_f32<T>(elem(to_f32<T>(sv.v[k]), to_f32<T>(vv.v[k])));
    st4(shadow, j, sv);
  }
};

struct ZeroOp {
  static constexpr bool vec = true;
  template <typename T>
  __device__ void apply(const MTMeta& m, int ti, long j) const {
    reinterpret_cast<T*>(m.a[ti])[j] = from_f32<T>(0.f);
  }
  template <typename T>
  __device__ void apply4(const MTMeta& m, int ti, long j) const {
    V4<T> z;
#pragma unroll
    for (int k = 0; k < 4; ++k) z.v[k] = from_f32<T>(0.f);
    st4(reinterpret_cast<T*>(m.a[ti]), j, z);
  }
};

struct ScaleOp {
  static constexpr bool vec = true;
  const float* norm;                  // device scalar
  float clip;
  template <typename T>
  __device__ void apply(const MTMeta& m, int ti, long j) const {
    T* t = reinterpret_cast<T*>(m.a[ti]);
    float n = *norm;
    float scale = clip / fmaxf(n, clip);
    t[j] = from_f32<T>(to_f32<T>(t[j]) * scale);
  }
  template <typename T>
  __device__ void apply4(const MTMeta& m, int ti, long j) const {
    T* t = reinterpret_cast<T*>(m.a[ti]);
    const float scale = clip / fmaxf(*norm, clip);
    V4<T> v = ld4(t, j);
#pragma unroll
    for (int k = 0; k < 4; ++k)
      v.v[k] = from_f32<T>(to_f32<T>(v.v[k]) * scale);
    st4(t, j, v);
  }
};

// --- host-side launch plumbing ---------------------------------------------

template <typename OP>
static void mt_launch(const std::vector<at::Tensor>& a,
                      const std::vector<at::Tensor>& b,
                      const std::vector<at::Tensor>& c,
                      const std::vector<at::Tensor>& d,
                      const std::vector<float>& e, OP op,
                      const std::vector<at::Tensor>& f = {}) {
  TORCH_CHECK(!a.empty());
  TORCH_CHECK(f.empty() || f.size() == a.size(),
              "mt_launch: masters list size mismatch");
  auto stream = at::cuda::getCurrentCUDAStream();
  size_t i = 0;
  while (i < a.size()) {
    MTMeta meta;
    meta.n = 0;
    meta.cum[0] = 0;
    meta.vcum[0] = 0;
    bool aligned = true;
    auto chk = [&aligned](const void* p) {
      if (p && (reinterpret_cast<uintptr_t>(p) & 15)) aligned = false;
    };
    while (i < a.size() && meta.n < MT_MAX) {
      int k = meta.n;
      meta.a[k] = a[i].data_ptr();
      meta.b[k] = b.empty() ? nullptr : b[i].data_ptr();
      meta.c[k] = c.empty() ? nullptr : c[i].data_ptr();
      meta.d[k] = d.empty() ? nullptr : d[i].data_ptr();
      meta.f[k] = f.empty() ? nullptr : f[i].data_ptr();
      meta.e[k] = e.empty() ? 0.f : e[i];
      meta.cum[k + 1] = meta.cum[k] + a[i].numel();
      meta.vcum[k + 1] = meta.vcum[k] + a[i].numel() / 4;
      chk(meta.a[k]);
      chk(meta.b[k]);
      chk(meta.c[k]);
      chk(meta.d[k]);
      chk(meta.f[k]);
      meta.n++;
      i++;
    }
    const long total = meta.cum[meta.n];
    const long vtotal = meta.vcum[meta.n];
    AT_DISPATCH_FLOATING_TYPES_AND2(
        at::kBFloat16, at::kHalf, a[0].scalar_type(), "mt_launch", [&] {
          if constexpr (OP::vec) {
            if (aligned) {
              mt_kernel_v4<scalar_t, OP>
                  <<<px_grid(vtotal > 0 ? vtotal : 1), PX_BLOCK, 0,
                     stream>>>(meta, op);
              HIP_CHECK_KERNEL();
              return;
            }
          }
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
    // the launch dispatches ONE element type from a[0] and reinterprets
    // every pointer with it — a mismatched-dtype grad (e.g. an fp32
    // reduce feeding a bf16 param) must fail loudly, not corrupt
    TORCH_CHECK(a[i].scalar_type() == b[i].scalar_type(),
                name, ": dtype mismatch at tensor ", i, " (param ",
                a[i].scalar_type(), " vs grad ", b[i].scalar_type(), ")");
  }
}

// Optional fp32 master weights: pass one fp32 tensor per (bf16) param, or
// an empty list for the plain path.
static void check_masters(const std::vector<at::Tensor>& params,
                          const std::vector<at::Tensor>& masters,
                          const char* name) {
  if (masters.empty()) return;
  TORCH_CHECK(masters.size() == params.size(), name,
              ": masters list size mismatch");
  for (size_t i = 0; i < masters.size(); ++i) {
    TORCH_CHECK(masters[i].scalar_type() == at::kFloat &&
                    masters[i].numel() == params[i].numel(),
                name, ": master ", i, " must be fp32 with the param's "
                "numel");
    TORCH_CHECK(masters[i].is_non_overlapping_and_dense(), name,
                ": master ", i, " must be dense");
  }
}

void multi_tensor_sgd(std::vector<at::Tensor> params,
                      std::vector<at::Tensor> grads, double lr, double wd,
                      std::vector<at::Tensor> masters) {
  check_contig(params, "mt_sgd");
  check_same_layout(params, grads, "mt_sgd");
  check_masters(params, masters, "mt_sgd");
  mt_launch(params, grads, {}, {}, {}, SGDOp{(float)lr, (float)wd},
            masters);
}

void multi_tensor_momentum(std::vector<at::Tensor> params,
                           std::vector<at::Tensor> grads,
                           std::vector<at::Tensor> bufs, double lr,
                           double mom, double wd, bool nesterov,
                           std::vector<at::Tensor> masters) {
  check_contig(params, "mt_momentum");
  check_same_layout(params, grads, "mt_momentum");
  check_same_layout(params, bufs, "mt_momentum");
  check_masters(params, masters, "mt_momentum");
  mt_launch(params, grads, bufs, {}, {},
            MomentumOp{(float)lr, (float)mom, (float)wd, nesterov},
            masters);
}

void multi_tensor_adagrad(std::vector<at::Tensor> params,
                          std::vector<at::Tensor> grads,
                          std::vector<at::Tensor> accums, double lr,
                          double eps, std::vector<at::Tensor> masters) {
  check_contig(params, "mt_adagrad");
  check_same_layout(params, grads, "mt_adagrad");
  check_masters(params, masters, "mt_adagrad");
  mt_launch(params, grads, accums, {}, {},
            AdagradOp{(float)lr, (float)eps}, masters);
}

void multi_tensor_adam(std::vector<at::Tensor> params,
                       std::vector<at::Tensor> grads,
                       std::vector<at::Tensor> ms,
                       std::vector<at::Tensor> vs, std::vector<double> steps,
                       double lr, double b1, double b2, double eps,
                       double wd, std::vector<at::Tensor> masters) {
  check_contig(params, "mt_adam");
  check_same_layout(params, grads, "mt_adam");
  check_masters(params, masters, "mt_adam");
  std::vector<float> e(steps.begin(), steps.end());
  mt_launch(params, grads, ms, vs, e,
            AdamOp{(float)lr, (float)b1, (float)b2, (float)eps, (float)wd,
                   nullptr},
            masters);
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
                           double wd, std::vector<at::Tensor> masters) {
  check_contig(params, "mt_adam_dev");
  check_same_layout(params, grads, "mt_adam_dev");
  check_masters(params, masters, "mt_adam_dev");
  TORCH_CHECK(step_dev.is_cuda() &&
                  step_dev.scalar_type() == at::kFloat &&
                  step_dev.numel() == 1,
              "mt_adam_dev: step_dev must be a 0-dim fp32 CUDA tensor");
  mt_launch(params, grads, ms, vs, {},
            AdamOp{(float)lr, (float)b1, (float)b2, (float)eps, (float)wd,
                   step_dev.data_ptr<float>()},
            masters);
}

void multi_tensor_rmsprop(std::vector<at::Tensor> params,
                          std::vector<at::Tensor> grads,
                          std::vector<at::Tensor> sqs,
                          std::vector<at::Tensor> bufs, double lr,
                          double decay, double mom, double eps,
                          std::vector<at::Tensor> masters) {
  check_contig(params, "mt_rmsprop");
  check_same_layout(params, grads, "mt_rmsprop");
  check_masters(params, masters, "mt_rmsprop");
  mt_launch(params, grads, sqs, bufs, {},
            RMSPropOp{(float)lr, (float)decay, (float)mom, (float)eps},
            masters);
}

void multi_tensor_ema(std::vector<at::Tensor> shadows,
                      std::vector<at::Tensor> values, double decay) {
  check_contig(shadows, "mt_ema");
  mt_launch(shadows, values, {}, {}, {}, EMAOp{(float)decay});
}

// Fused zero of a tensor list (one launch per 32 tensors) — replaces the
// per-tensor zero_() storm at the top of a captured step.
void multi_tensor_zero(std::vector<at::Tensor> tensors) {
  check_contig(tensors, "mt_zero");
  mt_launch(tensors, {}, {}, {}, {}, ZeroOp{});
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

// Vectorized variant: 4-element chunks (vcum index space) + scalar tails.
template <typename T>
__global__ void mt_sqsum_kernel_v4(MTMeta meta, float* out) {
  __shared__ float scratch[PX_BLOCK / WAVE];
  const long vtotal = meta.vcum[meta.n];
  float acc = 0.f;
  for (long vidx = (long)blockIdx.x * blockDim.x + threadIdx.x;
       vidx < vtotal; vidx += (long)gridDim.x * blockDim.x) {
    const int ti = mt_find_in(meta.vcum, meta.n, vidx);
    const long j = 4 * (vidx - meta.vcum[ti]);
    V4<T> v = ld4(reinterpret_cast<const T*>(meta.a[ti]), j);
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float f = to_f32<T>(v.v[k]);
      acc += f * f;
    }
  }
  if (blockIdx.x == 0 && threadIdx.x < 4 * (unsigned)meta.n) {
    const int ti = threadIdx.x >> 2;
    const long numel = meta.cum[ti + 1] - meta.cum[ti];
    const long j = (numel & ~3L) + (threadIdx.x & 3);
    if (j < numel) {
      float f = to_f32<T>(reinterpret_cast<const T*>(meta.a[ti])[j]);
      acc += f * f;
    }
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
    meta.vcum[0] = 0;
    bool aligned = true;
    while (i < tensors.size() && meta.n < MT_MAX) {
      int k = meta.n;
      meta.a[k] = tensors[i].data_ptr();
      meta.cum[k + 1] = meta.cum[k] + tensors[i].numel();
      meta.vcum[k + 1] = meta.vcum[k] + tensors[i].numel() / 4;
      if (reinterpret_cast<uintptr_t>(meta.a[k]) & 15) aligned = false;
      meta.n++;
      i++;
    }
    AT_DISPATCH_FLOATING_TYPES_AND2(
        at::kBFloat16, at::kHalf, tensors[0].scalar_type(), "mt_l2norm",
        [&] {
          if (aligned) {
            const long vt = meta.vcum[meta.n];
            mt_sqsum_kernel_v4<scalar_t>
                <<<px_grid(vt > 0 ? vt : 1), PX_BLOCK, 0, stream>>>(
                    meta, out.data_ptr<float>());
          } else {
            mt_sqsum_kernel<scalar_t>
                <<<px_grid(meta.cum[meta.n]), PX_BLOCK, 0, stream>>>(
                    meta, out.data_ptr<float>());
          }
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

// ``nvalid``: optional 0-dim int64 DEVICE scalar bounding the number of
// real rows — the sync-free padded dedup (ops/sparse.py
// unique_sum_rows_padded) emits fixed-shape id/row buffers whose tail is
// (id 0, zero rows); masking here avoids the tail's read-modify-write of
// row 0 racing a real id-0 update.  Each thread loads it once.
template <typename T>
__global__ void sparse_adagrad_kernel(T* __restrict__ p,
                                      float* __restrict__ acc,
                                      const int64_t* __restrict__ ids,
                                      const T* __restrict__ grows, long u,
                                      long d, float lr, float eps,
                                      const int64_t* __restrict__ nvalid) {
  const long nv = nvalid ? *nvalid : u;
  const long total = u * d;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long r = idx / d;
    if (r >= nv) continue;
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
                    at::Tensor grad_rows, double lr, double eps,
                    c10::optional<at::Tensor> nvalid) {
  TORCH_CHECK(param.is_cuda() && param.is_contiguous());
  TORCH_CHECK(accum.scalar_type() == at::kFloat);
  const long u = ids.numel();
  if (u == 0) return;
  const long d = param.dim() == 1 ? 1 : param.size(1);
  auto g = grad_rows.contiguous();
  const int64_t* nv = nullptr;
  if (nvalid.has_value()) {
    TORCH_CHECK(nvalid->scalar_type() == at::kLong && nvalid->is_cuda());
    nv = nvalid->data_ptr<int64_t>();
  }
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, param.scalar_type(), "sparse_adagrad", [&] {
        sparse_adagrad_kernel<scalar_t>
            <<<px_grid(u * d), PX_BLOCK, 0, stream>>>(
                reinterpret_cast<scalar_t*>(param.data_ptr()),
                accum.data_ptr<float>(), ids.data_ptr<int64_t>(),
                reinterpret_cast<const scalar_t*>(g.data_ptr()), u, d,
                (float)lr, (float)eps, nv);
        HIP_CHECK_KERNEL();
      });
}

template <typename T>
__global__ void sparse_sgd_kernel(T* __restrict__ p,
                                  const int64_t* __restrict__ ids,
                                  const T* __restrict__ grows, long u, long d,
                                  float lr,
                                  const int64_t* __restrict__ nvalid) {
  const long nv = nvalid ? *nvalid : u;
  const long total = u * d;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long r = idx / d;
    if (r >= nv) continue;
    const long c = idx - r * d;
    const long row = ids[r];
    p[row * d + c] = from_f32<T>(to_f32<T>(p[row * d + c]) -
                                 lr * to_f32<T>(grows[r * d + c]));
  }
}

void sparse_sgd(at::Tensor param, at::Tensor ids, at::Tensor grad_rows,
                double lr, c10::optional<at::Tensor> nvalid) {
  TORCH_CHECK(param.is_cuda() && param.is_contiguous());
  const long u = ids.numel();
  if (u == 0) return;
  const long d = param.dim() == 1 ? 1 : param.size(1);
  auto g = grad_rows.contiguous();
  const int64_t* nv = nullptr;
  if (nvalid.has_value()) {
    TORCH_CHECK(nvalid->scalar_type() == at::kLong && nvalid->is_cuda());
    nv = nvalid->data_ptr<int64_t>();
  }
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, param.scalar_type(), "sparse_sgd", [&] {
        sparse_sgd_kernel<scalar_t>
            <<<px_grid(u * d), PX_BLOCK, 0, stream>>>(
                reinterpret_cast<scalar_t*>(param.data_ptr()),
                ids.data_ptr<int64_t>(),
                reinterpret_cast<const scalar_t*>(g.data_ptr()), u, d,
                (float)lr, nv);
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

// Vectorized variants: require every tensor's numel % 4 == 0 (so the
// flat offsets stay 4-aligned) and 16B-aligned pointers — checked host
// side; no tails by construction.
template <typename TSrc, typename TDst>
__global__ void mt_pack_kernel_v4(MTMeta meta, TDst* __restrict__ flat) {
  const long vtotal = meta.vcum[meta.n];
  for (long vidx = (long)blockIdx.x * blockDim.x + threadIdx.x;
       vidx < vtotal; vidx += (long)gridDim.x * blockDim.x) {
    const int ti = mt_find_in(meta.vcum, meta.n, vidx);
    const long j = 4 * (vidx - meta.vcum[ti]);
    V4<TSrc> v = ld4(reinterpret_cast<const TSrc*>(meta.a[ti]), j);
    V4<TDst> o;
#pragma unroll
    for (int k = 0; k < 4; ++k)
      o.v[k] = from_f32<TDst>(to_f32<TSrc>(v.v[k]));
    st4(flat, meta.cum[ti] + j, o);
  }
}

template <typename TSrc, typename TDst>
__global__ void mt_unpack_kernel_v4(MTMeta meta,
                                    const TSrc* __restrict__ flat,
                                    float scale) {
  const long vtotal = meta.vcum[meta.n];
  for (long vidx = (long)blockIdx.x * blockDim.x + threadIdx.x;
       vidx < vtotal; vidx += (long)gridDim.x * blockDim.x) {
    const int ti = mt_find_in(meta.vcum, meta.n, vidx);
    const long j = 4 * (vidx - meta.vcum[ti]);
    V4<TSrc> v = ld4(flat, meta.cum[ti] + j);
    V4<TDst> o;
#pragma unroll
    for (int k = 0; k < 4; ++k)
      o.v[k] = from_f32<TDst>(to_f32<TSrc>(v.v[k]) * scale);
    st4(reinterpret_cast<TDst*>(meta.a[ti]), j, o);
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
    bool vec_ok = (base % 4 == 0);
    meta.vcum[0] = 0;
    while (i < tensors.size() && meta.n < MT_MAX) {
      int k = meta.n;
      meta.a[k] = tensors[i].data_ptr();
      meta.cum[k + 1] = meta.cum[k] + tensors[i].numel();
      meta.vcum[k + 1] = meta.vcum[k] + tensors[i].numel() / 4;
      if (tensors[i].numel() % 4 != 0 ||
          (reinterpret_cast<uintptr_t>(meta.a[k]) & 15))
        vec_ok = false;
      meta.n++;
      i++;
    }
    if (reinterpret_cast<uintptr_t>(flat.data_ptr()) & 15) vec_ok = false;
    const long total = meta.cum[meta.n];
    AT_DISPATCH_FLOATING_TYPES_AND2(
        at::kBFloat16, at::kHalf, tensors[0].scalar_type(), "mt_pack_t",
        [&] {
          using TP = scalar_t;
          AT_DISPATCH_FLOATING_TYPES_AND2(
              at::kBFloat16, at::kHalf, flat.scalar_type(), "mt_pack_f",
              [&] {
                using TF = scalar_t;
                const long vt = meta.vcum[meta.n];
                if (PACK && vec_ok) {
                  mt_pack_kernel_v4<TP, TF>
                      <<<px_grid(vt > 0 ? vt : 1), PX_BLOCK, 0, stream>>>(
                          meta,
                          reinterpret_cast<TF*>(flat.data_ptr()) + base);
                } else if (PACK) {
                  mt_pack_kernel<TP, TF>
                      <<<px_grid(total), PX_BLOCK, 0, stream>>>(
                          meta,
                          reinterpret_cast<TF*>(flat.data_ptr()) + base);
                } else if (vec_ok) {
                  mt_unpack_kernel_v4<TF, TP>
                      <<<px_grid(vt > 0 ? vt : 1), PX_BLOCK, 0, stream>>>(
                          meta,
                          reinterpret_cast<const TF*>(flat.data_ptr())
                              + base,
                          (float)scale);
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
