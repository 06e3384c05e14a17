#include "hip/hip_runtime.h"
// NHWC BatchNorm2d training fwd/bwd with optional fused ReLU
// (SURVEY §2.4 worklist: "BatchNorm fwd/bwd (decay .999) + ReLU").
//
// Layout: channels_last activations viewed [M = N*H*W, C] with C
// contiguous; C is a power of two (ResNet: 64..2048).
// Stats reduction: each 256-thread block sweeps a row slab; a thread owns
// channels {tid mod C} (C <= 256) or {tid, tid+256, ...} (C > 256), so
// per-element accumulation is PRIVATE registers; one global atomicAdd per
// (block, owned channel) merges the partials — no LDS traffic, no
// per-element atomics.  Backward reuses the structure for the two
// per-channel sums; the fused-ReLU mask is recomputed from
// (xhat*gamma+beta > 0), so nothing extra is saved.

#include <torch/extension.h>
#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

#define BN_MAXCPT 8  // max channels per thread (C <= 2048 at 256 threads)

template <typename T, bool XHAT>
__global__ __launch_bounds__(PX_BLOCK)
void bn_colsum_kernel(const T* __restrict__ x,
                                 const T* __restrict__ dy,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ invstd,
                                 const float* __restrict__ gamma,
                                 const float* __restrict__ beta,
                                 float* __restrict__ out1,
                                 float* __restrict__ out2, long M, int C,
                                 bool relu) {
  // XHAT=false: out1 = sum(x), out2 = sum(x^2)            (forward stats)
  // XHAT=true:  out1 = sum(dy_eff), out2 = sum(dy_eff * xhat)   (backward)
  const int tid = threadIdx.x;
  const int cpt = C > (int)blockDim.x ? C / (int)blockDim.x : 1;
  int own_c[BN_MAXCPT];
  float acc1[BN_MAXCPT], acc2[BN_MAXCPT];
#pragma unroll
  for (int k = 0; k < BN_MAXCPT; ++k) {
    own_c[k] = (C > (int)blockDim.x) ? (tid + k * (int)blockDim.x)
                                     : (tid & (C - 1));
    acc1[k] = 0.f;
    acc2[k] = 0.f;
  }
  const long total = M * (long)C;
  for (long idx = (long)blockIdx.x * blockDim.x + tid; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const int c_here = (int)(idx & (C - 1));
    // which of this thread's owned slots is c_here?  With the mapping
    // above and blockDim=256, idx stride keeps c_here == own_c[k] for
    // k = (idx/blockDim) % cpt when C > blockDim, and own_c[0] always
    // when C <= blockDim (stride % C == 0).
    const int k = (C > (int)blockDim.x)
                      ? (int)((idx / blockDim.x) % cpt)
                      : 0;
    if (XHAT) {
      const float xh = (to_f32<T>(x[idx]) - mean[c_here]) *
                       invstd[c_here];
      float dyv = to_f32<T>(dy[idx]);
      if (relu) {
        const float y = xh * gamma[c_here] + beta[c_here];
        dyv = y > 0.f ? dyv : 0.f;
      }
      acc1[k] += dyv;
      acc2[k] += dyv * xh;
    } else {
      const float v = to_f32<T>(x[idx]);
      acc1[k] += v;
      acc2[k] += v * v;
    }
  }
  // block-level LDS accumulation first: ONE global atomic per
  // (block, channel), not per thread — 512 blocks x C addresses instead
  // of 131k threads hammering C cells
  __shared__ float lsum1[256 * BN_MAXCPT];
  __shared__ float lsum2[256 * BN_MAXCPT];
  for (int c = tid; c < C; c += (int)blockDim.x) {
    lsum1[c] = 0.f;
    lsum2[c] = 0.f;
  }
  __syncthreads();
#pragma unroll
  for (int k = 0; k < BN_MAXCPT; ++k) {
    if (k < cpt) {
      atomicAdd(&lsum1[own_c[k]], acc1[k]);
      atomicAdd(&lsum2[own_c[k]], acc2[k]);
    }
  }
  __syncthreads();
  for (int c = tid; c < C; c += (int)blockDim.x) {
    if (lsum1[c] != 0.f || lsum2[c] != 0.f) {
      atomicAdd(&out1[c], lsum1[c]);
      atomicAdd(&out2[c], lsum2[c]);
    }
  }
}

__global__ void bn_finalize_kernel(const float* __restrict__ s1,
                                   const float* __restrict__ s2,
                                   float* __restrict__ mean,
                                   float* __restrict__ invstd,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var, long M,
                                   int C, float eps, float momentum) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const float m = s1[c] / (float)M;
  float var = s2[c] / (float)M - m * m;
  var = fmaxf(var, 0.f);
  mean[c] = m;
  invstd[c] = rsqrtf(var + eps);
  if (running_mean != nullptr) {
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * m;
    const float unbiased = var * (float)M / (float)(M > 1 ? M - 1 : 1);
    running_var[c] = (1.f - momentum) * running_var[c] +
                     momentum * unbiased;
  }
}

template <typename T>
__global__ void bn_norm_kernel(const T* __restrict__ x, T* __restrict__ y,
                               const float* __restrict__ mean,
                               const float* __restrict__ invstd,
                               const float* __restrict__ gamma,
                               const float* __restrict__ beta, long M,
                               int C, bool relu) {
  const long total = M * (long)C;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const int c = (int)(idx & (C - 1));
    float v = (to_f32<T>(x[idx]) - mean[c]) * invstd[c] * gamma[c] +
              beta[c];
    if (relu) v = fmaxf(v, 0.f);
    y[idx] = from_f32<T>(v);
  }
}

template <typename T>
__global__ void bn_dx_kernel(const T* __restrict__ x,
                             const T* __restrict__ dy, T* __restrict__ dx,
                             const float* __restrict__ mean,
                             const float* __restrict__ invstd,
                             const float* __restrict__ gamma,
                             const float* __restrict__ beta,
                             const float* __restrict__ sum_dy,
                             const float* __restrict__ sum_dy_xhat, long M,
                             int C, bool relu) {
  const long total = M * (long)C;
  const float invM = 1.f / (float)M;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const int c = (int)(idx & (C - 1));
    const float xh = (to_f32<T>(x[idx]) - mean[c]) * invstd[c];
    float dyv = to_f32<T>(dy[idx]);
    if (relu) {
      const float yv = xh * gamma[c] + beta[c];
      dyv = yv > 0.f ? dyv : 0.f;
    }
    const float g = gamma[c] * invstd[c];
    dx[idx] = from_f32<T>(
        g * (dyv - (sum_dy[c] + xh * sum_dy_xhat[c]) * invM));
  }
}

std::vector<at::Tensor> batchnorm_fwd(const at::Tensor& x2d,
                                      const at::Tensor& gamma,
                                      const at::Tensor& beta,
                                      c10::optional<at::Tensor> run_mean,
                                      c10::optional<at::Tensor> run_var,
                                      double momentum, double eps,
                                      bool relu) {
  TORCH_CHECK(x2d.is_cuda() && x2d.dim() == 2 && x2d.is_contiguous());
  const long M = x2d.size(0);
  const int C = (int)x2d.size(1);
  TORCH_CHECK((C & (C - 1)) == 0 && C <= 256 * BN_MAXCPT,
              "batchnorm: C must be a power of two <= 2048");
  auto f32 = x2d.options().dtype(at::kFloat);
  auto s1 = at::zeros({C}, f32);
  auto s2 = at::zeros({C}, f32);
  auto mean = at::empty({C}, f32);
  auto invstd = at::empty({C}, f32);
  auto y = at::empty_like(x2d);
  auto g32 = gamma.to(at::kFloat).contiguous();
  auto b32 = beta.to(at::kFloat).contiguous();
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, x2d.scalar_type(), "bn_fwd", [&] {
       hipLaunchKernelGGL(( bn_colsum_kernel<scalar_t, false>)
            , dim3(px_grid(M * C, PX_BLOCK, 512)), dim3(PX_BLOCK), 0, stream, 
                reinterpret_cast<const scalar_t*>(x2d.data_ptr()), nullptr,
                nullptr, nullptr, nullptr, nullptr, s1.data_ptr<float>(),
                s2.data_ptr<float>(), M, C, false);
        HIP_CHECK_KERNEL();
       hipLaunchKernelGGL(( bn_finalize_kernel), dim3((C + 255) / 256), dim3(256), 0, stream, 
            s1.data_ptr<float>(), s2.data_ptr<float>(),
            mean.data_ptr<float>(), invstd.data_ptr<float>(),
            run_mean.has_value() ? run_mean->data_ptr<float>() : nullptr,
            run_var.has_value() ? run_var->data_ptr<float>() : nullptr, M,
            C, (float)eps, (float)momentum);
        HIP_CHECK_KERNEL();
       hipLaunchKernelGGL(( bn_norm_kernel<scalar_t>), dim3(px_grid(M * C)), dim3(PX_BLOCK), 0, stream, 
            reinterpret_cast<const scalar_t*>(x2d.data_ptr()),
            reinterpret_cast<scalar_t*>(y.data_ptr()),
            mean.data_ptr<float>(), invstd.data_ptr<float>(),
            g32.data_ptr<float>(), b32.data_ptr<float>(), M, C, relu);
        HIP_CHECK_KERNEL();
      });
  return {y, mean, invstd};
}

std::vector<at::Tensor> batchnorm_bwd(const at::Tensor& dy2d,
                                      const at::Tensor& x2d,
                                      const at::Tensor& mean,
                                      const at::Tensor& invstd,
                                      const at::Tensor& gamma,
                                      const at::Tensor& beta, bool relu) {
  const long M = x2d.size(0);
  const int C = (int)x2d.size(1);
  auto f32 = x2d.options().dtype(at::kFloat);
  auto sum_dy = at::zeros({C}, f32);
  auto sum_dy_xhat = at::zeros({C}, f32);
  auto dx = at::empty_like(x2d);
  auto g32 = gamma.to(at::kFloat).contiguous();
  auto b32 = beta.to(at::kFloat).contiguous();
  auto dyc = dy2d.contiguous();
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, x2d.scalar_type(), "bn_bwd", [&] {
       hipLaunchKernelGGL(( bn_colsum_kernel<scalar_t, true>)
            , dim3(px_grid(M * C, PX_BLOCK, 512)), dim3(PX_BLOCK), 0, stream, 
                reinterpret_cast<const scalar_t*>(x2d.data_ptr()),
                reinterpret_cast<const scalar_t*>(dyc.data_ptr()),
                mean.data_ptr<float>(), invstd.data_ptr<float>(),
                g32.data_ptr<float>(), b32.data_ptr<float>(),
                sum_dy.data_ptr<float>(), sum_dy_xhat.data_ptr<float>(),
                M, C, relu);
        HIP_CHECK_KERNEL();
       hipLaunchKernelGGL(( bn_dx_kernel<scalar_t>), dim3(px_grid(M * C)), dim3(PX_BLOCK), 0, stream, 
            reinterpret_cast<const scalar_t*>(x2d.data_ptr()),
            reinterpret_cast<const scalar_t*>(dyc.data_ptr()),
            reinterpret_cast<scalar_t*>(dx.data_ptr()),
            mean.data_ptr<float>(), invstd.data_ptr<float>(),
            g32.data_ptr<float>(), b32.data_ptr<float>(),
            sum_dy.data_ptr<float>(), sum_dy_xhat.data_ptr<float>(), M, C,
            relu);
        HIP_CHECK_KERNEL();
      });
  // dgamma = sum_dy_xhat, dbeta = sum_dy (per-channel sums of the
  // (relu-masked) upstream grad)
  return {dx, sum_dy_xhat, sum_dy};
}
