// LayerNorm forward/backward (SURVEY §2.4 worklist).
//
// Forward: one block per row (grid-stride): two-pass-free mean/var via a
// single sum/sumsq sweep, wave64 shuffle + cross-wave LDS reduction, then
// the normalize+affine write.  Saves (mean, rstd) fp32 per row.
// Backward: per-row dx with the two standard row statistics
// (mean(dy*w), mean(dy*w*xhat)).  dw/db: for d <= 4096 each block
// accumulates fp32 partials in LDS (2*d floats) over its rows and flushes
// ONCE per column with a global fp32 atomicAdd (the BN-colsum recipe —
// grid*d atomics total, no per-row global RMW), then one tiny kernel
// casts to the weight dtype.  Larger d falls back to a [grid, D]
// workspace + column-sum second kernel.

#include <torch/extension.h>
#include <ATen/ATen.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

static inline bool ln_aligned16(const at::Tensor& t) {
  return (reinterpret_cast<uintptr_t>(t.data_ptr()) & 15) == 0;
}


template <typename T, bool VEC>
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
    if (VEC) {  // d % 4 == 0, 16B-aligned rows (checked on host)
      for (long c = threadIdx.x * 4; c < d; c += (long)blockDim.x * 4) {
        PxV4<T> v = px_ld4(xr, c);
#pragma unroll
        for (int k = 0; k < 4; ++k) {
          float f = to_f32<T>(v.v[k]);
          s += f;
          ss += f * f;
        }
      }
    } else {
      for (long c = threadIdx.x; c < d; c += blockDim.x) {
        float v = to_f32<T>(xr[c]);
        s += v;
        ss += v * v;
      }
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
    if (VEC) {
      for (long c = threadIdx.x * 4; c < d; c += (long)blockDim.x * 4) {
        PxV4<T> xv = px_ld4(xr, c), wv = px_ld4(w, c), bv = px_ld4(b, c);
        PxV4<T> yv;
#pragma unroll
        for (int k = 0; k < 4; ++k) {
          float v = (to_f32<T>(xv.v[k]) - m) * r;
          yv.v[k] = from_f32<T>(v * to_f32<T>(wv.v[k]) +
                                to_f32<T>(bv.v[k]));
        }
        px_st4(yr, c, yv);
      }
    } else {
      for (long c = threadIdx.x; c < d; c += blockDim.x) {
        float v = (to_f32<T>(xr[c]) - m) * r;
        yr[c] = from_f32<T>(v * to_f32<T>(w[c]) + to_f32<T>(b[c]));
      }
    }
  }
}

std::vector<at::Tensor> layernorm_fwd(const at::Tensor& x,
                                      const at::Tensor& w,
                                      const at::Tensor& b, double eps) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  TORCH_CHECK(w.scalar_type() == x.scalar_type() &&
              b.scalar_type() == x.scalar_type(),
              "layernorm_fwd: weight/bias dtype must match x (the kernel "
              "reads them with x's element type)");
  const long n = x.size(0), d = x.size(1);
  auto y = at::empty_like(x);
  auto mean = at::empty({n}, x.options().dtype(at::kFloat));
  auto rstd = at::empty({n}, x.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  int grid = (int)(n < PX_MAX_GRID ? n : PX_MAX_GRID);
  auto wc = w.contiguous();
  auto bc = b.contiguous();
  const bool vec = (d % 4 == 0) && ln_aligned16(x) && ln_aligned16(wc) &&
                   ln_aligned16(bc);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, x.scalar_type(), "layernorm_fwd", [&] {
        auto launch = [&](auto vtag) {
          layernorm_fwd_kernel<scalar_t, decltype(vtag)::value>
              <<<grid, PX_BLOCK, 0, stream>>>(
                  reinterpret_cast<const scalar_t*>(x.data_ptr()),
                  reinterpret_cast<const scalar_t*>(wc.data_ptr()),
                  reinterpret_cast<const scalar_t*>(bc.data_ptr()),
                  reinterpret_cast<scalar_t*>(y.data_ptr()),
                  mean.data_ptr<float>(), rstd.data_ptr<float>(), n, d,
                  (float)eps);
        };
        if (vec) launch(std::true_type{});
        else launch(std::false_type{});
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

// LDS-resident dw/db accumulation (d <= LN_LDS_MAX_D columns): one atomic
// flush per (block, column) instead of a per-row global read-modify-write.
// The flush stripes over LN_NBUF partial buffers (blockIdx % LN_NBUF) —
// with one buffer, 512 blocks hammering the same d cells serialize ~512
// deep per cell; striping cuts the contention chain to grid/LN_NBUF.
// The cast kernel folds the stripes.
#define LN_LDS_MAX_D 4096
#define LN_NBUF 16

template <typename T, bool VEC>
__global__ void layernorm_bwd_lds_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const T* __restrict__ w, const float* __restrict__ mean,
    const float* __restrict__ rstd, T* __restrict__ dx,
    float* __restrict__ dw32, float* __restrict__ db32, long n, long d) {
  __shared__ float scratch[PX_BLOCK / WAVE];
  extern __shared__ float smem[];  // [2 * d]: dw then db partials
  float* sdw = smem;
  float* sdb = smem + d;
  for (long c = threadIdx.x; c < d; c += blockDim.x) {
    sdw[c] = 0.f;
    sdb[c] = 0.f;
  }
  __syncthreads();
  for (long row = blockIdx.x; row < n; row += gridDim.x) {
    const T* dyr = dy + row * d;
    const T* xr = x + row * d;
    const float m = mean[row];
    const float r = rstd[row];
    float s1 = 0.f, s2 = 0.f;
    if (VEC) {
      for (long c = threadIdx.x * 4; c < d; c += (long)blockDim.x * 4) {
        PxV4<T> dyv4 = px_ld4(dyr, c), xv4 = px_ld4(xr, c),
                wv4 = px_ld4(w, c);
#pragma unroll
        for (int k = 0; k < 4; ++k) {
          float dyv = to_f32<T>(dyv4.v[k]);
          float xh = (to_f32<T>(xv4.v[k]) - m) * r;
          float dyw = dyv * to_f32<T>(wv4.v[k]);
          s1 += dyw;
          s2 += dyw * xh;
          sdw[c + k] += dyv * xh;
          sdb[c + k] += dyv;
        }
      }
    } else {
      for (long c = threadIdx.x; c < d; c += blockDim.x) {
        float dyv = to_f32<T>(dyr[c]);
        float xh = (to_f32<T>(xr[c]) - m) * r;
        float dyw = dyv * to_f32<T>(w[c]);
        s1 += dyw;
        s2 += dyw * xh;
        sdw[c] += dyv * xh;  // LDS: same thread owns column c all rows
        sdb[c] += dyv;
      }
    }
    s1 = block_reduce_sum(s1, scratch) / d;
    s2 = block_reduce_sum(s2, scratch) / d;
    T* dxr = dx + row * d;
    if (VEC) {
      for (long c = threadIdx.x * 4; c < d; c += (long)blockDim.x * 4) {
        PxV4<T> dyv4 = px_ld4(dyr, c), xv4 = px_ld4(xr, c),
                wv4 = px_ld4(w, c);
        PxV4<T> dxv;
#pragma unroll
        for (int k = 0; k < 4; ++k) {
          float xh = (to_f32<T>(xv4.v[k]) - m) * r;
          float dyw = to_f32<T>(dyv4.v[k]) * to_f32<T>(wv4.v[k]);
          dxv.v[k] = from_f32<T>(r * (dyw - s1 - xh * s2));
        }
        px_st4(dxr, c, dxv);
      }
    } else {
      for (long c = threadIdx.x; c < d; c += blockDim.x) {
        float xh = (to_f32<T>(xr[c]) - m) * r;
        float dyw = to_f32<T>(dyr[c]) * to_f32<T>(w[c]);
        dxr[c] = from_f32<T>(r * (dyw - s1 - xh * s2));
      }
    }
  }
  __syncthreads();
  // per-block partial rows, plain stores (nothing pre-zeroed, no atomic
  // contention; a wave-per-column reduce folds them + casts)
  float* dwb = dw32 + (long)blockIdx.x * 2 * d;
  for (long c = threadIdx.x; c < d; c += blockDim.x) {
    dwb[c] = sdw[c];
    dwb[d + c] = sdb[c];
  }
  (void)db32;
}

// Reduce the [nblk][2d] partials (one wave per output column) and cast
// to the weight dtype in one launch.
template <typename T>
__global__ __launch_bounds__(256)
void fold_cast2_kernel(const float* __restrict__ part, int nblk,
                       T* __restrict__ a, T* __restrict__ b, long d) {
  const long c = blockIdx.x * 4 + (threadIdx.x >> 6);
  if (c >= d) return;
  const int lane = threadIdx.x & 63;
  float sa = 0.f, sb = 0.f;
  for (int k = lane; k < nblk; k += 64) {
    sa += part[(long)k * 2 * d + c];
    sb += part[(long)k * 2 * d + d + c];
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    sa += __shfl_down(sa, off, 64);
    sb += __shfl_down(sb, off, 64);
  }
  if (lane != 0) return;
  a[c] = from_f32<T>(sa);
  b[c] = from_f32<T>(sb);
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
  TORCH_CHECK(w.scalar_type() == x.scalar_type(),
              "layernorm_bwd: weight dtype must match x");
  const long n = x.size(0), d = x.size(1);
  auto dx = at::empty_like(x);
  auto dw = at::empty_like(w);
  auto db = at::empty_like(w);
  auto stream = at::cuda::getCurrentCUDAStream();
  if (d <= LN_LDS_MAX_D) {
    long g = n / 8;
    if (g > 512) g = 512;
    if (g < 16) g = 16;
    if (g > n) g = n > 0 ? n : 1;
    const int grid = (int)g;
    // per-block partials, nothing to pre-zero
    auto ws = at::empty({(long)grid * 2 * d},
                        x.options().dtype(at::kFloat));
    auto dw32 = ws;
    auto db32 = ws;
    const size_t lds = 2 * (size_t)d * sizeof(float);
    auto wc = w.contiguous();
    const bool vec = (d % 4 == 0) && ln_aligned16(dy) &&
                     ln_aligned16(x) && ln_aligned16(wc);
    AT_DISPATCH_FLOATING_TYPES_AND2(
        at::kBFloat16, at::kHalf, x.scalar_type(), "layernorm_bwd", [&] {
          auto launch = [&](auto vtag) {
            layernorm_bwd_lds_kernel<scalar_t, decltype(vtag)::value>
                <<<grid, PX_BLOCK, lds, stream>>>(
                    reinterpret_cast<const scalar_t*>(dy.data_ptr()),
                    reinterpret_cast<const scalar_t*>(x.data_ptr()),
                    reinterpret_cast<const scalar_t*>(wc.data_ptr()),
                    mean.data_ptr<float>(), rstd.data_ptr<float>(),
                    reinterpret_cast<scalar_t*>(dx.data_ptr()),
                    dw32.data_ptr<float>(), db32.data_ptr<float>(), n, d);
          };
          if (vec) launch(std::true_type{});
          else launch(std::false_type{});
          HIP_CHECK_KERNEL();
          fold_cast2_kernel<scalar_t>
              <<<(int)((d + 3) / 4), 256, 0, stream>>>(
              dw32.data_ptr<float>(), grid,
              reinterpret_cast<scalar_t*>(dw.data_ptr()),
              reinterpret_cast<scalar_t*>(db.data_ptr()), d);
          HIP_CHECK_KERNEL();
        });
    return {dx, dw, db};
  }
  const int grid = (int)(n < 256 ? (n > 0 ? n : 1) : 256);
  auto dw_part = at::zeros({grid, d}, x.options().dtype(at::kFloat));
  auto db_part = at::zeros({grid, d}, x.options().dtype(at::kFloat));
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, x.scalar_type(), "layernorm_bwd", [&] {
        layernorm_bwd_kernel<scalar_t><<<grid, PX_BLOCK, 0, stream>>>(
            reinterpret_cast<const scalar_t*>(dy.data_ptr()),
            reinterpret_cast<const scalar_t*>(x.data_ptr()),
            reinterpret_cast<const scalar_t*>(w.contiguous().data_ptr()),
            mean.data_ptr<float>(), rstd.data_ptr<float>(),
            reinterpret_cast<scalar_t*>(dx.data_ptr()),
            dw_part.data_ptr<float>(), db_part.data_ptr<float>(), n, d);
        HIP_CHECK_KERNEL();
        colsum_kernel<scalar_t><<<px_grid(d), PX_BLOCK, 0, stream>>>(
            dw_part.data_ptr<float>(),
            reinterpret_cast<scalar_t*>(dw.data_ptr()), grid, d);
        colsum_kernel<scalar_t><<<px_grid(d), PX_BLOCK, 0, stream>>>(
            db_part.data_ptr<float>(),
            reinterpret_cast<scalar_t*>(db.data_ptr()), grid, d);
        HIP_CHECK_KERNEL();
      });
  return {dx, dw, db};
}
