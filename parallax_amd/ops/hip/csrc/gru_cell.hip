// Fused GRU gate pointwise math (skip_thoughts hot op: reference
// examples/skip_thoughts GRU encoder/decoders).
//
// Given xg = x@W_x + b [B, 3H] (layout r|z|n) and hg = h@W_h [B, 3H]:
//   r = sigmoid(xg_r + hg_r); z = sigmoid(xg_z + hg_z)
//   n = tanh(xg_n + r * hg_n); h' = (1-z)*n + z*h
// in ONE kernel; backward emits d_xg [B,3H], d_hg [B,3H], d_hprev.

#include <torch/extension.h>
#include <ATen/ATen.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

__device__ __forceinline__ float gru_sigmoid(float x) {
  return 1.f / (1.f + __expf(-x));
}

template <typename T>
__global__ void gru_gates_fwd_kernel(
    const T* __restrict__ xg, const T* __restrict__ hg,
    const T* __restrict__ h_prev, T* __restrict__ h_out,
    float* __restrict__ r_s, float* __restrict__ z_s,
    float* __restrict__ n_t, float* __restrict__ hgn_s, long bh,
    long hsz) {
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < bh;
       idx += (long)gridDim.x * blockDim.x) {
    const long b = idx / hsz;
    const long h = idx - b * hsz;
    const long g0 = b * 3 * hsz + h;
    const float r = gru_sigmoid(to_f32<T>(xg[g0]) + to_f32<T>(hg[g0]));
    const float z = gru_sigmoid(to_f32<T>(xg[g0 + hsz]) +
                                to_f32<T>(hg[g0 + hsz]));
    const float hgn = to_f32<T>(hg[g0 + 2 * hsz]);
    const float n = tanhf(to_f32<T>(xg[g0 + 2 * hsz]) + r * hgn);
    r_s[idx] = r;
    z_s[idx] = z;
    n_t[idx] = n;
    hgn_s[idx] = hgn;
    h_out[idx] =
        from_f32<T>((1.f - z) * n + z * to_f32<T>(h_prev[idx]));
  }
}

std::vector<at::Tensor> gru_gates_fwd(const at::Tensor& xg,
                                      const at::Tensor& hg,
                                      const at::Tensor& h_prev) {
  TORCH_CHECK(xg.is_cuda() && xg.dim() == 2 && xg.is_contiguous());
  const long B = xg.size(0);
  const long H = xg.size(1) / 3;
  auto hgc = hg.contiguous();
  auto hp = h_prev.contiguous();
  auto h_out = at::empty({B, H}, xg.options());
  auto f32 = xg.options().dtype(at::kFloat);
  auto r_s = at::empty({B, H}, f32);
  auto z_s = at::empty({B, H}, f32);
  auto n_t = at::empty({B, H}, f32);
  auto hgn_s = at::empty({B, H}, f32);
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, xg.scalar_type(), "gru_gates_fwd", [&] {
        gru_gates_fwd_kernel<scalar_t>
            <<<px_grid(B * H), PX_BLOCK, 0, stream>>>(
                reinterpret_cast<const scalar_t*>(xg.data_ptr()),
                reinterpret_cast<const scalar_t*>(hgc.data_ptr()),
                reinterpret_cast<const scalar_t*>(hp.data_ptr()),
                reinterpret_cast<scalar_t*>(h_out.data_ptr()),
                r_s.data_ptr<float>(), z_s.data_ptr<float>(),
                n_t.data_ptr<float>(), hgn_s.data_ptr<float>(), B * H, H);
        HIP_CHECK_KERNEL();
      });
  return {h_out, r_s, z_s, n_t, hgn_s};
}

template <typename T>
__global__ void gru_gates_bwd_kernel(
    const T* __restrict__ dh, const T* __restrict__ h_prev,
    const float* __restrict__ r_s, const float* __restrict__ z_s,
    const float* __restrict__ n_t, const float* __restrict__ hgn_s,
    T* __restrict__ d_xg, T* __restrict__ d_hg, T* __restrict__ d_hprev,
    long bh, long hsz) {
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < bh;
       idx += (long)gridDim.x * blockDim.x) {
    const long b = idx / hsz;
    const long h = idx - b * hsz;
    const long g0 = b * 3 * hsz + h;
    const float dhv = to_f32<T>(dh[idx]);
    const float r = r_s[idx], z = z_s[idx], n = n_t[idx];
    const float hgn = hgn_s[idx];
    const float hp = to_f32<T>(h_prev[idx]);
    const float dn = dhv * (1.f - z);
    const float dz = dhv * (hp - n);
    const float dn_pre = dn * (1.f - n * n);
    const float dr = dn_pre * hgn;
    const float dr_pre = dr * r * (1.f - r);
    const float dz_pre = dz * z * (1.f - z);
    d_xg[g0] = from_f32<T>(dr_pre);
    d_xg[g0 + hsz] = from_f32<T>(dz_pre);
    d_xg[g0 + 2 * hsz] = from_f32<T>(dn_pre);
    d_hg[g0] = from_f32<T>(dr_pre);
    d_hg[g0 + hsz] = from_f32<T>(dz_pre);
    d_hg[g0 + 2 * hsz] = from_f32<T>(dn_pre * r);
    d_hprev[idx] = from_f32<T>(dhv * z);
  }
}

std::vector<at::Tensor> gru_gates_bwd(
    const at::Tensor& dh, const at::Tensor& h_prev, const at::Tensor& r_s,
    const at::Tensor& z_s, const at::Tensor& n_t, const at::Tensor& hgn_s) {
  const long B = dh.size(0), H = dh.size(1);
  auto d_xg = at::empty({B, 3 * H}, dh.options());
  auto d_hg = at::empty({B, 3 * H}, dh.options());
  auto d_hprev = at::empty({B, H}, dh.options());
  auto hp = h_prev.contiguous();
  auto dhc = dh.contiguous();
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, dh.scalar_type(), "gru_gates_bwd", [&] {
        gru_gates_bwd_kernel<scalar_t>
            <<<px_grid(B * H), PX_BLOCK, 0, stream>>>(
                reinterpret_cast<const scalar_t*>(dhc.data_ptr()),
                reinterpret_cast<const scalar_t*>(hp.data_ptr()),
                r_s.data_ptr<float>(), z_s.data_ptr<float>(),
                n_t.data_ptr<float>(), hgn_s.data_ptr<float>(),
                reinterpret_cast<scalar_t*>(d_xg.data_ptr()),
                reinterpret_cast<scalar_t*>(d_hg.data_ptr()),
                reinterpret_cast<scalar_t*>(d_hprev.data_ptr()), B * H, H);
        HIP_CHECK_KERNEL();
      });
  return {d_xg, d_hg, d_hprev};
}
