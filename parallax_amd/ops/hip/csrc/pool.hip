// NHWC pooling kernels (SURVEY §2.4 worklist row "max/avg pool";
// reference convnet_builder.py mpool/apool + ResNet's stem
// MaxPool2d(3, stride 2, pad 1) and global average pool).
//
// Channels-last, bf16, C % 8 == 0: every access is a 16-byte short8
// over 8 consecutive channels (guide G13).  Max-pool backward is
// GATHER-based (each input position scans the <= 2x2 output windows
// that can contain it and sums the dy where it was the argmax) — no
// atomics, deterministic.

#include <torch/extension.h>
#include <ATen/ATen.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short p_short8;

__device__ __forceinline__ float p_bf(short v) {
  union { short s; __hip_bfloat16 b; } u;
  u.s = v;
  return __bfloat162float(u.b);
}
__device__ __forceinline__ short p_fb(float f) {
  union { short s; __hip_bfloat16 b; } u;
  u.b = __float2bfloat16(f);
  return u.s;
}

// ---- max pool 3x3 stride 2 pad 1 fwd: also writes per-element argmax
// (window slot 0..8, int8) for the backward gather --------------------------
__global__ __launch_bounds__(256)
void maxpool3x3s2_fwd_kernel(const short* __restrict__ x,
                             short* __restrict__ y,
                             char* __restrict__ arg, int N, int H, int W,
                             int C, int OH, int OW) {
  const long total = (long)N * OH * OW * (C >> 3);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const int c8 = (int)(i % (C >> 3));
    long t = i / (C >> 3);
    const int ox = (int)(t % OW);
    t /= OW;
    const int oy = (int)(t % OH);
    const int n = (int)(t / OH);
    const int cg = c8 << 3;
    float best[8];
    char slot[8];
#pragma unroll
    for (int q = 0; q < 8; ++q) {
      best[q] = -3e38f;
      slot[q] = 0;
    }
    const int y0 = oy * 2 - 1, x0 = ox * 2 - 1;
#pragma unroll
    for (int dy = 0; dy < 3; ++dy) {
      const int yy = y0 + dy;
      if (yy < 0 || yy >= H) continue;
#pragma unroll
      for (int dx = 0; dx < 3; ++dx) {
        const int xx = x0 + dx;
        if (xx < 0 || xx >= W) continue;
        const p_short8 v = *reinterpret_cast<const p_short8*>(
            &x[(((long)n * H + yy) * W + xx) * C + cg]);
#pragma unroll
        for (int q = 0; q < 8; ++q) {
          const float f = p_bf(v[q]);
          if (f > best[q]) {
            best[q] = f;
            slot[q] = (char)(dy * 3 + dx);
          }
        }
      }
    }
    p_short8 o;
#pragma unroll
    for (int q = 0; q < 8; ++q) o[q] = p_fb(best[q]);
    const long obase = (((long)n * OH + oy) * OW + ox) * C + cg;
    *reinterpret_cast<p_short8*>(&y[obase]) = o;
#pragma unroll
    for (int q = 0; q < 8; ++q) arg[obase + q] = slot[q];
  }
}

// backward: dx[n,y,x,c] = sum over output windows (oy,ox) with
// argmax slot pointing at (y,x)
__global__ __launch_bounds__(256)
void maxpool3x3s2_bwd_kernel(const short* __restrict__ dyp,
                             const char* __restrict__ arg,
                             short* __restrict__ dx, int N, int H, int W,
                             int C, int OH, int OW) {
  const long total = (long)N * H * W * (C >> 3);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const int c8 = (int)(i % (C >> 3));
    long t = i / (C >> 3);
    const int xx = (int)(t % W);
    t /= W;
    const int yy = (int)(t % H);
    const int n = (int)(t / H);
    const int cg = c8 << 3;
    float acc[8] = {};
    // windows containing (yy, xx): oy*2-1 <= yy <= oy*2+1
    const int oy_lo = max(0, (yy + 1 - 2 + 1) / 2), oy_hi =
        min(OH - 1, (yy + 1) / 2);
    const int ox_lo = max(0, (xx + 1 - 2 + 1) / 2), ox_hi =
        min(OW - 1, (xx + 1) / 2);
    for (int oy = oy_lo; oy <= oy_hi; ++oy) {
      const int dy = yy - (oy * 2 - 1);
      if (dy < 0 || dy > 2) continue;
      for (int ox = ox_lo; ox <= ox_hi; ++ox) {
        const int dxx = xx - (ox * 2 - 1);
        if (dxx < 0 || dxx > 2) continue;
        const char want = (char)(dy * 3 + dxx);
        const long obase = (((long)n * OH + oy) * OW + ox) * C + cg;
        const p_short8 g = *reinterpret_cast<const p_short8*>(
            &dyp[obase]);
#pragma unroll
        for (int q = 0; q < 8; ++q) {
          if (arg[obase + q] == want) acc[q] += p_bf(g[q]);
        }
      }
    }
    p_short8 o;
#pragma unroll
    for (int q = 0; q < 8; ++q) o[q] = p_fb(acc[q]);
    *reinterpret_cast<p_short8*>(
        &dx[(((long)n * H + yy) * W + xx) * C + cg]) = o;
  }
}

// ---- global average pool fwd/bwd ---------------------------------------
__global__ __launch_bounds__(256)
void gavgpool_fwd_kernel(const short* __restrict__ x,
                         float* __restrict__ y, int N, int HW, int C) {
  // one block per image; threads own channel groups
  const int n = blockIdx.x;
  for (int cg = threadIdx.x * 8; cg < C; cg += blockDim.x * 8) {
    float acc[8] = {};
    for (int p = 0; p < HW; ++p) {
      const p_short8 v = *reinterpret_cast<const p_short8*>(
          &x[((long)n * HW + p) * C + cg]);
#pragma unroll
      for (int q = 0; q < 8; ++q) acc[q] += p_bf(v[q]);
    }
    const float inv = 1.f / (float)HW;
#pragma unroll
    for (int q = 0; q < 8; ++q) y[(long)n * C + cg + q] = acc[q] * inv;
  }
}

__global__ __launch_bounds__(256)
void gavgpool_bwd_kernel(const float* __restrict__ dyp,
                         short* __restrict__ dx, int N, int HW, int C) {
  const long total = (long)N * HW * (C >> 3);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const int c8 = (int)(i % (C >> 3));
    const long t = i / (C >> 3);
    const int n = (int)(t / HW);
    const int cg = c8 << 3;
    const float inv = 1.f / (float)HW;
    p_short8 o;
#pragma unroll
    for (int q = 0; q < 8; ++q) {
      o[q] = p_fb(dyp[(long)n * C + cg + q] * inv);
    }
    *reinterpret_cast<p_short8*>(&dx[i * 8]) = o;
  }
}

static int pool_grid(long work) {
  long b = (work + 255) / 256;
  if (b > 2048) b = 2048;
  if (b < 1) b = 1;
  return (int)b;
}

std::vector<at::Tensor> maxpool3x3s2_fwd(const at::Tensor& x2d, long N,
                                         long H, long W, long C) {
  TORCH_CHECK(x2d.is_cuda() && x2d.scalar_type() == at::kBFloat16 &&
              x2d.is_contiguous() && C % 8 == 0);
  const long OH = (H + 1) / 2, OW = (W + 1) / 2;
  auto y = at::empty({N * OH * OW * C}, x2d.options());
  auto arg = at::empty({N * OH * OW * C},
                       x2d.options().dtype(at::kChar));
  auto stream = at::cuda::getCurrentCUDAStream();
  const long work = N * OH * OW * (C / 8);
  maxpool3x3s2_fwd_kernel<<<pool_grid(work), 256, 0, stream>>>(
      reinterpret_cast<const short*>(x2d.data_ptr()),
      reinterpret_cast<short*>(y.data_ptr()),
      reinterpret_cast<char*>(arg.data_ptr()), (int)N, (int)H, (int)W,
      (int)C, (int)OH, (int)OW);
  HIP_CHECK_KERNEL();
  return {y, arg};
}

at::Tensor maxpool3x3s2_bwd(const at::Tensor& dy, const at::Tensor& arg,
                            long N, long H, long W, long C) {
  const long OH = (H + 1) / 2, OW = (W + 1) / 2;
  auto dx = at::empty({N * H * W * C}, dy.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  const long work = N * H * W * (C / 8);
  maxpool3x3s2_bwd_kernel<<<pool_grid(work), 256, 0, stream>>>(
      reinterpret_cast<const short*>(dy.data_ptr()),
      reinterpret_cast<const char*>(arg.data_ptr()),
      reinterpret_cast<short*>(dx.data_ptr()), (int)N, (int)H, (int)W,
      (int)C, (int)OH, (int)OW);
  HIP_CHECK_KERNEL();
  return dx;
}

at::Tensor gavgpool_fwd(const at::Tensor& x2d, long N, long HW, long C) {
  TORCH_CHECK(x2d.is_cuda() && x2d.scalar_type() == at::kBFloat16 &&
              x2d.is_contiguous() && C % 8 == 0);
  auto y = at::empty({N, C}, x2d.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  gavgpool_fwd_kernel<<<(int)N, 256, 0, stream>>>(
      reinterpret_cast<const short*>(x2d.data_ptr()),
      y.data_ptr<float>(), (int)N, (int)HW, (int)C);
  HIP_CHECK_KERNEL();
  return y;
}

at::Tensor gavgpool_bwd(const at::Tensor& dy, long N, long HW, long C) {
  auto dyf = dy.to(at::kFloat).contiguous();
  auto dx = at::empty({N * HW * C},
                      dy.options().dtype(at::kBFloat16));
  auto stream = at::cuda::getCurrentCUDAStream();
  const long work = N * HW * (C / 8);
  gavgpool_bwd_kernel<<<pool_grid(work), 256, 0, stream>>>(
      dyf.data_ptr<float>(), reinterpret_cast<short*>(dx.data_ptr()),
      (int)N, (int)HW, (int)C);
  HIP_CHECK_KERNEL();
  return dx;
}
