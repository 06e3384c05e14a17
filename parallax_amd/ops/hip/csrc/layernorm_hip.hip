#include "hip/hip_runtime.h"
// LayerNorm forward/backward (SURVEY §2.4 worklist).
//
// Forward: one block per row (grid-stride): two-pass-free mean/var via a
// single sum/sumsq sweep, wave64 shuffle + cross-wave LDS reduction, then
// the normalize+affine write.  Saves (mean, rstd) fp32 per row.
// Backward: per-row dx with the two standard row statistics
// (mean(dy*w), mean(dy*w*xhat)); dw/db accumulated as per-block partials
// in a [grid, D] fp32 workspace, reduced by a second kernel — no global
// atomics on the hot path.

#include <torch/extension.h>
#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

template <typename T>
__global__ void layernorm_fwd_kernel(const T* __restrict__ x,
                                     const T* __restrict__ w,
                                     const T* __restrict__ b,
                                     T* __restrict__ y,
                                     float* __restrict__ mean,
                                     float* __restrict__ rstd, long n,
                                     long d, float eps) {
  __shared__ float scratch[PX_BLOCK / WAVE];
  for (long row = blockIdx.x; row < n; row += gridDim.x) {
    const T* xr = x + row * d;
    float s = 0.f, ss = 0.f;
    for (long c = threadIdx.x; c < d; c += blockDim.x) {
      float v = to_f32<T>(xr[c]);
      s += v;
      ss += v * v;
    }
    s = block_reduce_sum(s, scratch);
    ss = block_reduce_sum(ss, scratch);
    const float m = s / d;
    float var = ss / d - m * m;
    var = fmaxf(var, 0.f);
    const float r = rsqrtf(var + eps);
    if (threadIdx.x == 0) {
      mean[row] = m;
      rstd[row] = r;
    }
    T* yr = y + row * d;
    for (long c = threadIdx.x; c < d; c += blockDim.x) {
      float v = (to_f32<T>(xr[c]) - m) * r;
      yr[c] = from_f32<T>(v * to_f32<T>(w[c]) + to_f32<T>(b[c]));
    }
  }
}

std::vector<at::Tensor> layernorm_fwd(const at::Tensor& x,
                                      const at::Tensor& w,
                                      const at::Tensor& b, double eps) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  const long n = x.size(0), d = x.size(1);
  auto y = at::empty_like(x);
  auto mean = at::empty({n}, x.options().dtype(at::kFloat));
  auto rstd = at::empty({n}, x.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  int grid = (int)(n < PX_MAX_GRID ? n : PX_MAX_GRID);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, x.scalar_type(), "layernorm_fwd", [&] {
       hipLaunchKernelGGL(( layernorm_fwd_kernel<scalar_t>), dim3(grid), dim3(PX_BLOCK), 0, stream, 
            reinterpret_cast<const scalar_t*>(x.data_ptr()),
            reinterpret_cast<const scalar_t*>(w.contiguous().data_ptr()),
            reinterpret_cast<const scalar_t*>(b.contiguous().data_ptr()),
            reinterpret_cast<scalar_t*>(y.data_ptr()),
            mean.data_ptr<float>(), rstd.data_ptr<float>(), n, d,
            (float)eps);
        HIP_CHECK_KERNEL();
      });
  return {y, mean, rstd};
}

template <typename T>
__global__ void layernorm_bwd_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const T* __restrict__ w, const float* __restrict__ mean,
    const float* __restrict__ rstd, T* __restrict__ dx,
    float* __restrict__ dw_part, float* __restrict__ db_part, long n,
    long d) {
  __shared__ float scratch[PX_BLOCK / WAVE];
  float* dwp = dw_part + (long)blockIdx.x * d;
  float* dbp = db_part + (long)blockIdx.x * d;
  for (long row = blockIdx.x; row < n; row += gridDim.x) {
    const T* dyr = dy + row * d;
    const T* xr = x + row * d;
    const float m = mean[row];
    const float r = rstd[row];
    float s1 = 0.f, s2 = 0.f;
    for (long c = threadIdx.x; c < d; c += blockDim.x) {
      float xh = (to_f32<T>(xr[c]) - m) * r;
      float dyw = to_f32<T>(dyr[c]) * to_f32<T>(w[c]);
      s1 += dyw;
      s2 += dyw * xh;
      // per-block dw/db partials
      dwp[c] += to_f32<T>(dyr[c]) * xh;
      dbp[c] += to_f32<T>(dyr[c]);
    }
    s1 = block_reduce_sum(s1, scratch) / d;
    s2 = block_reduce_sum(s2, scratch) / d;
    T* dxr = dx + row * d;
    for (long c = threadIdx.x; c < d; c += blockDim.x) {
      float xh = (to_f32<T>(xr[c]) - m) * r;
      float dyw = to_f32<T>(dyr[c]) * to_f32<T>(w[c]);
      dxr[c] = from_f32<T>(r * (dyw - s1 - xh * s2));
    }
  }
}

template <typename T>
__global__ void colsum_kernel(const float* __restrict__ part, T* __restrict__
                              out, long rows, long d) {
  for (long c = (long)blockIdx.x * blockDim.x + threadIdx.x; c < d;
       c += (long)gridDim.x * blockDim.x) {
    float acc = 0.f;
    for (long r = 0; r < rows; ++r) acc += part[r * d + c];
    out[c] = from_f32<T>(acc);
  }
}

std::vector<at::Tensor> layernorm_bwd(const at::Tensor& dy,
                                      const at::Tensor& x,
                                      const at::Tensor& w,
                                      const at::Tensor& mean,
                                      const at::Tensor& rstd) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  const long n = x.size(0), d = x.size(1);
  auto dx = at::empty_like(x);
  auto dw = at::empty_like(w);
  auto db = at::empty_like(w);
  const int grid = (int)(n < 256 ? (n > 0 ? n : 1) : 256);
  auto dw_part = at::zeros({grid, d}, x.options().dtype(at::kFloat));
  auto db_part = at::zeros({grid, d}, x.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, x.scalar_type(), "layernorm_bwd", [&] {
       hipLaunchKernelGGL(( layernorm_bwd_kernel<scalar_t>), dim3(grid), dim3(PX_BLOCK), 0, stream, 
            reinterpret_cast<const scalar_t*>(dy.data_ptr()),
            reinterpret_cast<const scalar_t*>(x.data_ptr()),
            reinterpret_cast<const scalar_t*>(w.contiguous().data_ptr()),
            mean.data_ptr<float>(), rstd.data_ptr<float>(),
            reinterpret_cast<scalar_t*>(dx.data_ptr()),
            dw_part.data_ptr<float>(), db_part.data_ptr<float>(), n, d);
        HIP_CHECK_KERNEL();
       hipLaunchKernelGGL(( colsum_kernel<scalar_t>), dim3(px_grid(d)), dim3(PX_BLOCK), 0, stream, 
            dw_part.data_ptr<float>(),
            reinterpret_cast<scalar_t*>(dw.data_ptr()), grid, d);
       hipLaunchKernelGGL(( colsum_kernel<scalar_t>), dim3(px_grid(d)), dim3(PX_BLOCK), 0, stream, 
            db_part.data_ptr<float>(),
            reinterpret_cast<scalar_t*>(db.data_ptr()), grid, d);
        HIP_CHECK_KERNEL();
      });
  return {dx, dw, db};
}
