// NHWC BatchNorm2d training fwd/bwd with fused ReLU and fused residual
// add (SURVEY §2.4 worklist: "BatchNorm fwd/bwd (decay .999) + ReLU";
// reference convnet_builder.py:428-501 conv+BN+ReLU chains and the
// ResNet bottleneck's bn3 + shortcut-add + ReLU).
//
// These are pure HBM-streaming kernels; the design rules are guide G13
// (bf16 loads vectorized 16 B/lane — scalar bf16 is ~2x slower) and a
// FIXED channel-group-per-thread mapping so per-channel accumulators
// and scale/shift constants live in registers, never LDS:
//
//   * activations are channels_last views [M = N*H*W, C], C % 8 == 0,
//     C <= 2048;
//   * a thread owns ONE group of 8 consecutive channels; threads_per_row
//     = C/8; a 256-thread block covers floor(256/tpr) rows per sweep and
//     grid-strides over rows — every load is a 16-byte short8;
//   * stats reduce: per-thread fp32 acc[8] -> LDS combine -> one global
//     atomicAdd per (block, channel);
//   * finalize precomputes scale = gamma*invstd and shift = beta -
//     mean*scale, so the normalize pass is one fma per element (+ the
//     optional residual addend read and ReLU) — y = relu(x*scale + shift
//     [+ id]);
//   * backward reads the SAVED OUTPUT y for the ReLU mask (with the add
//     fused, the mask depends on the post-add value, which cannot be
//     recomputed from x alone) and optionally writes the identity
//     branch's gradient dy' = dy * (y > 0) in the same pass.

#include <torch/extension.h>
#include <ATen/ATen.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8v;

template <typename T>
__device__ __forceinline__ float bn_elt(const short8v v, int k);
template <>
__device__ __forceinline__ float bn_elt<at::BFloat16>(const short8v v,
                                                      int k) {
  union { short s; __hip_bfloat16 b; } u;
  u.s = v[k];
  return __bfloat162float(u.b);
}
template <>
__device__ __forceinline__ float bn_elt<at::Half>(const short8v v, int k) {
  union { short s; __half h; } u;
  u.s = v[k];
  return __half2float(u.h);
}

template <typename T>
__device__ __forceinline__ short bn_pack(float f);
template <>
__device__ __forceinline__ short bn_pack<at::BFloat16>(float f) {
  union { short s; __hip_bfloat16 b; } u;
  u.b = __float2bfloat16(f);
  return u.s;
}
template <>
__device__ __forceinline__ short bn_pack<at::Half>(float f) {
  union { short s; __half h; } u;
  u.h = __float2half(f);
  return u.s;
}

// ---- forward stats: s1[c] = sum x, s2[c] = sum x^2 ----------------------
template <typename T>
__global__ __launch_bounds__(256)
void bn_stats_kernel(const T* __restrict__ x, float* __restrict__ s1,
                     float* __restrict__ s2, long M, int C) {
  const int tpr = C >> 3;                    // thread-groups per row
  const int rpb = 256 / tpr;                 // rows per block sweep
  const int tid = threadIdx.x;
  const int rsub = tid / tpr;                // row slot within the block
  const int cg = (tid - rsub * tpr) << 3;    // first owned channel
  float a1[8] = {}, a2[8] = {};
  if (rsub < rpb) {
    const long rstep = (long)gridDim.x * rpb;
    long r = (long)blockIdx.x * rpb + rsub;
    // 2 rows in flight per thread (independent loads hide latency)
    for (; r + rstep < M; r += 2 * rstep) {
      const short8v v0 = *(const short8v*)(x + r * C + cg);
      const short8v v1 = *(const short8v*)(x + (r + rstep) * C + cg);
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        const float f0 = bn_elt<T>(v0, k);
        const float f1 = bn_elt<T>(v1, k);
        a1[k] += f0 + f1;
        a2[k] += f0 * f0 + f1 * f1;
      }
    }
    for (; r < M; r += rstep) {
      const short8v v = *(const short8v*)(x + r * C + cg);
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        const float f = bn_elt<T>(v, k);
        a1[k] += f;
        a2[k] += f * f;
      }
    }
  }
  __shared__ float ls[2 * 2048];
  for (int c = tid; c < 2 * C; c += 256) ls[c] = 0.f;
  __syncthreads();
  if (rsub < rpb) {
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      atomicAdd(&ls[cg + k], a1[k]);
      atomicAdd(&ls[C + cg + k], a2[k]);
    }
  }
  __syncthreads();
  // per-block partials, plain stores (no global atomics, nothing to
  // pre-zero): part[block][0..C) = sums, part[block][C..2C) = sumsq
  float* mine = s1 + (long)blockIdx.x * 2 * C;
  for (int c = tid; c < C; c += 256) {
    mine[c] = ls[c];
    mine[C + c] = ls[C + c];
  }
  (void)s2;
}

// ---- finalize: mean/invstd, running stats, scale/shift ------------------
// one WAVE per channel (the partials are L2-hot from the stats kernel;
// 64 lanes x few loads each + shuffle reduce, instead of one thread
// looping over every block's partial)
__global__ __launch_bounds__(256)
void bn_finalize_kernel(const float* __restrict__ part,
                                   int nblk,
                                   const float* __restrict__ gamma,
                                   const float* __restrict__ beta,
                                   float* __restrict__ mean,
                                   float* __restrict__ invstd,
                                   float* __restrict__ scale,
                                   float* __restrict__ shift,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var, long M,
                                   int C, float eps, float momentum) {
  const int c = blockIdx.x * 4 + (threadIdx.x >> 6);
  if (c >= C) return;
  const int lane = threadIdx.x & 63;
  float sum = 0.f, sumsq = 0.f;
  for (int b = lane; b < nblk; b += 64) {
    sum += part[(long)b * 2 * C + c];
    sumsq += part[(long)b * 2 * C + C + c];
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    sum += __shfl_down(sum, off, 64);
    sumsq += __shfl_down(sumsq, off, 64);
  }
  if (lane != 0) return;
  const float m = sum / (float)M;
  float var = sumsq / (float)M - m * m;
  var = fmaxf(var, 0.f);
  const float is = rsqrtf(var + eps);
  mean[c] = m;
  invstd[c] = is;
  const float sc = gamma[c] * is;
  scale[c] = sc;
  shift[c] = beta[c] - m * sc;
  if (running_mean != nullptr) {
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * m;
    const float unbiased = var * (float)M / (float)(M > 1 ? M - 1 : 1);
    running_var[c] = (1.f - momentum) * running_var[c] +
                     momentum * unbiased;
  }
}

// ---- normalize (+ residual add) (+ relu) --------------------------------
template <typename T, bool ADD, bool RELU>
__global__ __launch_bounds__(256)
void bn_norm_kernel(const T* __restrict__ x, const T* __restrict__ addend,
                    T* __restrict__ y, const float* __restrict__ scale,
                    const float* __restrict__ shift, long M, int C) {
  const int tpr = C >> 3;
  const int rpb = 256 / tpr;
  const int tid = threadIdx.x;
  const int rsub = tid / tpr;
  const int cg = (tid - rsub * tpr) << 3;
  if (rsub >= rpb) return;
  float sc[8], sh[8];
#pragma unroll
  for (int k = 0; k < 8; ++k) {
    sc[k] = scale[cg + k];
    sh[k] = shift[cg + k];
  }
  const long rstep = (long)gridDim.x * rpb;
  for (long r = (long)blockIdx.x * rpb + rsub; r < M; r += rstep) {
    const long base = r * C + cg;
    const short8v v = *(const short8v*)(x + base);
    short8v a;
    if (ADD) a = *(const short8v*)(addend + base);
    short8v o;
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float f = fmaf(bn_elt<T>(v, k), sc[k], sh[k]);
      if (ADD) f += bn_elt<T>(a, k);
      if (RELU) f = fmaxf(f, 0.f);
      o[k] = bn_pack<T>(f);
    }
    *(short8v*)(y + base) = o;
  }
}

// ---- backward stats: sum dy', sum dy'*xhat ------------------------------
// dy' = dy * (y > 0) when RELU else dy; xhat = (x - mean) * invstd.
template <typename T, bool RELU>
__global__ __launch_bounds__(256)
void bn_bstats_kernel(const T* __restrict__ x, const T* __restrict__ dy,
                      const T* __restrict__ y,
                      const float* __restrict__ mean,
                      const float* __restrict__ invstd,
                      float* __restrict__ s1, float* __restrict__ s2,
                      long M, int C) {
  const int tpr = C >> 3;
  const int rpb = 256 / tpr;
  const int tid = threadIdx.x;
  const int rsub = tid / tpr;
  const int cg = (tid - rsub * tpr) << 3;
  float a1[8] = {}, a2[8] = {};
  if (rsub < rpb) {
    float mn[8], is[8];
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      mn[k] = mean[cg + k];
      is[k] = invstd[cg + k];
    }
    const long rstep = (long)gridDim.x * rpb;
    long r = (long)blockIdx.x * rpb + rsub;
    for (; r + rstep < M; r += 2 * rstep) {
      const long b0 = r * C + cg;
      const long b1 = (r + rstep) * C + cg;
      const short8v xv0 = *(const short8v*)(x + b0);
      const short8v dv0 = *(const short8v*)(dy + b0);
      const short8v xv1 = *(const short8v*)(x + b1);
      const short8v dv1 = *(const short8v*)(dy + b1);
      short8v yv0, yv1;
      if (RELU) {
        yv0 = *(const short8v*)(y + b0);
        yv1 = *(const short8v*)(y + b1);
      }
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        float d0 = bn_elt<T>(dv0, k);
        float d1 = bn_elt<T>(dv1, k);
        if (RELU && !(bn_elt<T>(yv0, k) > 0.f)) d0 = 0.f;
        if (RELU && !(bn_elt<T>(yv1, k) > 0.f)) d1 = 0.f;
        a1[k] += d0 + d1;
        a2[k] += d0 * (bn_elt<T>(xv0, k) - mn[k]) * is[k] +
                 d1 * (bn_elt<T>(xv1, k) - mn[k]) * is[k];
      }
    }
    for (; r < M; r += rstep) {
      const long base = r * C + cg;
      const short8v xv = *(const short8v*)(x + base);
      const short8v dv = *(const short8v*)(dy + base);
      short8v yv;
      if (RELU) yv = *(const short8v*)(y + base);
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        float d = bn_elt<T>(dv, k);
        if (RELU && !(bn_elt<T>(yv, k) > 0.f)) d = 0.f;
        a1[k] += d;
        a2[k] += d * (bn_elt<T>(xv, k) - mn[k]) * is[k];
      }
    }
  }
  __shared__ float ls[2 * 2048];
  for (int c = tid; c < 2 * C; c += 256) ls[c] = 0.f;
  __syncthreads();
  if (rsub < rpb) {
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      atomicAdd(&ls[cg + k], a1[k]);
      atomicAdd(&ls[C + cg + k], a2[k]);
    }
  }
  __syncthreads();
  // per-block partials, plain stores (nothing to pre-zero)
  float* mine = s1 + (long)blockIdx.x * 2 * C;
  for (int c = tid; c < C; c += 256) {
    mine[c] = ls[c];
    mine[C + c] = ls[C + c];
  }
  (void)s2;
}

// reduce backward partials -> sum_dy / sum_dy_xhat
__global__ __launch_bounds__(256)
void bn_breduce_kernel(const float* __restrict__ part, int nblk,
                                  float* __restrict__ sum_dy,
                                  float* __restrict__ sum_dy_xhat, int C) {
  const int c = blockIdx.x * 4 + (threadIdx.x >> 6);
  if (c >= C) return;
  const int lane = threadIdx.x & 63;
  float s1 = 0.f, s2 = 0.f;
  for (int b = lane; b < nblk; b += 64) {
    s1 += part[(long)b * 2 * C + c];
    s2 += part[(long)b * 2 * C + C + c];
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    s1 += __shfl_down(s1, off, 64);
    s2 += __shfl_down(s2, off, 64);
  }
  if (lane != 0) return;
  sum_dy[c] = s1;
  sum_dy_xhat[c] = s2;
}

// ---- backward dx (+ identity-branch grad) -------------------------------
// dx = gamma*invstd * (dy' - sum_dy'/M - xhat * sum_dy_xhat/M)
// dxid = dy'   (gradient through the fused residual add)
template <typename T, bool RELU, bool WANT_DID>
__global__ __launch_bounds__(256)
void bn_dx_kernel(const T* __restrict__ x, const T* __restrict__ dy,
                  const T* __restrict__ y, T* __restrict__ dx,
                  T* __restrict__ dxid, const float* __restrict__ mean,
                  const float* __restrict__ invstd,
                  const float* __restrict__ gamma,
                  const float* __restrict__ sum_dy,
                  const float* __restrict__ sum_dy_xhat, long M, int C) {
  const int tpr = C >> 3;
  const int rpb = 256 / tpr;
  const int tid = threadIdx.x;
  const int rsub = tid / tpr;
  const int cg = (tid - rsub * tpr) << 3;
  if (rsub >= rpb) return;
  const float invM = 1.f / (float)M;
  float mn[8], is[8], g[8], k1[8], k2[8];
#pragma unroll
  for (int k = 0; k < 8; ++k) {
    mn[k] = mean[cg + k];
    is[k] = invstd[cg + k];
    g[k] = gamma[cg + k] * is[k];
    k1[k] = sum_dy[cg + k] * invM;
    k2[k] = sum_dy_xhat[cg + k] * invM;
  }
  const long rstep = (long)gridDim.x * rpb;
  for (long r = (long)blockIdx.x * rpb + rsub; r < M; r += rstep) {
    const long base = r * C + cg;
    const short8v xv = *(const short8v*)(x + base);
    const short8v dv = *(const short8v*)(dy + base);
    short8v yv;
    if (RELU) yv = *(const short8v*)(y + base);
    short8v o, oid;
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float d = bn_elt<T>(dv, k);
      if (RELU && !(bn_elt<T>(yv, k) > 0.f)) d = 0.f;
      const float xh = (bn_elt<T>(xv, k) - mn[k]) * is[k];
      o[k] = bn_pack<T>(g[k] * (d - k1[k] - xh * k2[k]));
      if (WANT_DID) oid[k] = bn_pack<T>(d);
    }
    *(short8v*)(dx + base) = o;
    if (WANT_DID) *(short8v*)(dxid + base) = oid;
  }
}

static int bn_grid(long M, int C) {
  const int rpb = 256 / (C >> 3);
  long blocks = (M + rpb - 1) / rpb;
  if (blocks > 512) blocks = 512;  // bounds the partial buffer; 2 blocks
  if (blocks < 1) blocks = 1;      // per CU saturate a streaming kernel
  return (int)blocks;
}

// stats/bstats grid: ALSO proportional to the tensor, so the per-block
// partial rows (written by the reduce, re-read by finalize) stay <= a
// few % of the input bytes — a fixed 512-block grid put a ~15 us floor
// under every small-layer BN (partials 512 x 2C fp32 vs a 3 MB input).
static int bn_grid_stats(long M, int C) {
  long blocks = M / 32;
  if (blocks > 512) blocks = 512;
  if (blocks < 16) blocks = 16;
  const int rpb = 256 / (C >> 3);
  long max_useful = (M + rpb - 1) / rpb;
  if (blocks > max_useful) blocks = max_useful;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

std::vector<at::Tensor> batchnorm_fwd(const at::Tensor& x2d,
                                      const at::Tensor& gamma,
                                      const at::Tensor& beta,
                                      c10::optional<at::Tensor> run_mean,
                                      c10::optional<at::Tensor> run_var,
                                      double momentum, double eps,
                                      bool relu,
                                      c10::optional<at::Tensor> addend) {
  TORCH_CHECK(x2d.is_cuda() && x2d.dim() == 2 && x2d.is_contiguous());
  const long M = x2d.size(0);
  const int C = (int)x2d.size(1);
  TORCH_CHECK(C % 8 == 0 && C <= 2048 && (C >> 3) <= 256,
              "batchnorm: need C % 8 == 0 and C <= 2048");
  TORCH_CHECK(x2d.scalar_type() == at::kBFloat16 ||
              x2d.scalar_type() == at::kHalf,
              "batchnorm: bf16/fp16 activations only");
  auto f32 = x2d.options().dtype(at::kFloat);
  const int grid = bn_grid(M, C);
  const int sgrid = bn_grid_stats(M, C);
  // one workspace: [sgrid][2C] partials + mean/invstd/scale/shift
  auto ws = at::empty({(long)sgrid * 2 * C + 4 * C}, f32);
  float* wsp = ws.data_ptr<float>();
  float* partp = wsp;
  float* meanp = wsp + (long)sgrid * 2 * C;
  float* invstdp = meanp + C;
  float* scp = invstdp + C;
  float* shp = scp + C;
  auto mean = ws.narrow(0, (long)sgrid * 2 * C, C);
  auto invstd = ws.narrow(0, (long)sgrid * 2 * C + C, C);
  auto y = at::empty_like(x2d);
  auto g32 = gamma.to(at::kFloat).contiguous();
  auto b32 = beta.to(at::kFloat).contiguous();
  const bool add = addend.has_value();
  if (add) {
    TORCH_CHECK(addend->is_contiguous() &&
                addend->sizes() == x2d.sizes() &&
                addend->scalar_type() == x2d.scalar_type());
  }
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, x2d.scalar_type(), "bn_fwd", [&] {
        if constexpr (std::is_same<scalar_t, float>::value ||
                      std::is_same<scalar_t, double>::value) {
          TORCH_CHECK(false, "batchnorm: bf16/fp16 only");
        } else {
          bn_stats_kernel<scalar_t><<<sgrid, 256, 0, stream>>>(
              reinterpret_cast<const scalar_t*>(x2d.data_ptr()),
              partp, nullptr, M, C);
          HIP_CHECK_KERNEL();
          bn_finalize_kernel<<<(C + 3) / 4, 256, 0, stream>>>(
              partp, sgrid, g32.data_ptr<float>(), b32.data_ptr<float>(),
              meanp, invstdp, scp, shp,
              run_mean.has_value() ? run_mean->data_ptr<float>() : nullptr,
              run_var.has_value() ? run_var->data_ptr<float>() : nullptr,
              M, C, (float)eps, (float)momentum);
          HIP_CHECK_KERNEL();
          const scalar_t* ap = add
              ? reinterpret_cast<const scalar_t*>(addend->data_ptr())
              : nullptr;
          const scalar_t* xp =
              reinterpret_cast<const scalar_t*>(x2d.data_ptr());
          scalar_t* yp = reinterpret_cast<scalar_t*>(y.data_ptr());
          if (add && relu) {
            bn_norm_kernel<scalar_t, true, true><<<grid, 256, 0, stream>>>(
                xp, ap, yp, scp, shp, M, C);
          } else if (add) {
            bn_norm_kernel<scalar_t, true, false><<<grid, 256, 0,
                                                    stream>>>(
                xp, ap, yp, scp, shp, M, C);
          } else if (relu) {
            bn_norm_kernel<scalar_t, false, true><<<grid, 256, 0,
                                                    stream>>>(
                xp, ap, yp, scp, shp, M, C);
          } else {
            bn_norm_kernel<scalar_t, false, false><<<grid, 256, 0,
                                                     stream>>>(
                xp, ap, yp, scp, shp, M, C);
          }
          HIP_CHECK_KERNEL();
        }
      });
  return {y, mean, invstd};
}

std::vector<at::Tensor> batchnorm_bwd(const at::Tensor& dy2d,
                                      const at::Tensor& x2d,
                                      const at::Tensor& y2d,
                                      const at::Tensor& mean,
                                      const at::Tensor& invstd,
                                      const at::Tensor& gamma, bool relu,
                                      bool want_dxid) {
  const long M = x2d.size(0);
  const int C = (int)x2d.size(1);
  auto f32 = x2d.options().dtype(at::kFloat);
  const int grid = bn_grid(M, C);
  const int sgrid = bn_grid_stats(M, C);
  auto ws = at::empty({(long)sgrid * 2 * C + 2 * C}, f32);
  float* partp = ws.data_ptr<float>();
  float* sdyp = partp + (long)sgrid * 2 * C;
  float* sdxp = sdyp + C;
  auto sum_dy = ws.narrow(0, (long)sgrid * 2 * C, C);
  auto sum_dy_xhat = ws.narrow(0, (long)sgrid * 2 * C + C, C);
  auto dx = at::empty_like(x2d);
  auto dxid = want_dxid ? at::empty_like(x2d) : at::Tensor();
  auto g32 = gamma.to(at::kFloat).contiguous();
  auto dyc = dy2d.contiguous();
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, x2d.scalar_type(), "bn_bwd", [&] {
        if constexpr (std::is_same<scalar_t, float>::value ||
                      std::is_same<scalar_t, double>::value) {
          TORCH_CHECK(false, "batchnorm: bf16/fp16 only");
        } else {
          const scalar_t* xp =
              reinterpret_cast<const scalar_t*>(x2d.data_ptr());
          const scalar_t* dp =
              reinterpret_cast<const scalar_t*>(dyc.data_ptr());
          const scalar_t* yp =
              reinterpret_cast<const scalar_t*>(y2d.data_ptr());
          if (relu) {
            bn_bstats_kernel<scalar_t, true><<<sgrid, 256, 0, stream>>>(
                xp, dp, yp, mean.data_ptr<float>(),
                invstd.data_ptr<float>(), partp, nullptr, M, C);
          } else {
            bn_bstats_kernel<scalar_t, false><<<sgrid, 256, 0, stream>>>(
                xp, dp, yp, mean.data_ptr<float>(),
                invstd.data_ptr<float>(), partp, nullptr, M, C);
          }
          HIP_CHECK_KERNEL();
          bn_breduce_kernel<<<(C + 3) / 4, 256, 0, stream>>>(
              partp, sgrid, sdyp, sdxp, C);
          HIP_CHECK_KERNEL();
          scalar_t* didp = want_dxid
              ? reinterpret_cast<scalar_t*>(dxid.data_ptr()) : nullptr;
          scalar_t* dxp = reinterpret_cast<scalar_t*>(dx.data_ptr());
          float* mp = mean.data_ptr<float>();
          float* ip = invstd.data_ptr<float>();
          float* gp = g32.data_ptr<float>();
          float* s1p = sdyp;
          float* s2p = sdxp;
          if (relu && want_dxid) {
            bn_dx_kernel<scalar_t, true, true><<<grid, 256, 0, stream>>>(
                xp, dp, yp, dxp, didp, mp, ip, gp, s1p, s2p, M, C);
          } else if (relu) {
            bn_dx_kernel<scalar_t, true, false><<<grid, 256, 0, stream>>>(
                xp, dp, yp, dxp, didp, mp, ip, gp, s1p, s2p, M, C);
          } else if (want_dxid) {
            bn_dx_kernel<scalar_t, false, true><<<grid, 256, 0, stream>>>(
                xp, dp, yp, dxp, didp, mp, ip, gp, s1p, s2p, M, C);
          } else {
            bn_dx_kernel<scalar_t, false, false><<<grid, 256, 0,
                                                   stream>>>(
                xp, dp, yp, dxp, didp, mp, ip, gp, s1p, s2p, M, C);
          }
          HIP_CHECK_KERNEL();
        }
      });
  // dgamma = sum_dy_xhat, dbeta = sum_dy (per-channel sums of the
  // relu-masked upstream grad)
  if (want_dxid) return {dx, sum_dy_xhat, sum_dy, dxid};
  return {dx, sum_dy_xhat, sum_dy};
}
