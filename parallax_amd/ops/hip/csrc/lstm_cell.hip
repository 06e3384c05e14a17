// Fused LSTM gate pointwise math (the LM1B hot op: reference
// language_model.py:51-62 cell arithmetic).
//
// Forward, given pre-activation gates [B, 4H] (layout i|g|f|o) and c_prev:
//   i = sigmoid(gi); g = tanh(gg); f = sigmoid(gf + forget_bias);
//   o = sigmoid(go); c = f*c_prev + i*g; h = o*tanh(c)
// in ONE kernel (the eager alternative is ~10 elementwise launches reading
// the [B, 4H] tensor repeatedly).  Saves the activated gates for backward.
//
// Backward: dgates [B, 4H] + dc_prev in one kernel.

#include <torch/extension.h>
#include <ATen/ATen.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

__device__ __forceinline__ float sigmoidf_(float x) {
  return 1.f / (1.f + __expf(-x));
}

template <typename T>
__global__ void lstm_gates_fwd_kernel(
    const T* __restrict__ gates, const T* __restrict__ c_prev,
    T* __restrict__ h_out, T* __restrict__ c_out, float* __restrict__ i_s,
    float* __restrict__ g_t, float* __restrict__ f_s,
    float* __restrict__ o_s, long bh, long hsz, float forget_bias) {
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < bh;
       idx += (long)gridDim.x * blockDim.x) {
    const long b = idx / hsz;
    const long h = idx - b * hsz;
    const long g0 = b * 4 * hsz + h;
    const float gi = to_f32<T>(gates[g0]);
    const float gg = to_f32<T>(gates[g0 + hsz]);
    const float gf = to_f32<T>(gates[g0 + 2 * hsz]);
    const float go = to_f32<T>(gates[g0 + 3 * hsz]);
    const float iv = sigmoidf_(gi);
    const float gv = tanhf(gg);
    const float fv = sigmoidf_(gf + forget_bias);
    const float ov = sigmoidf_(go);
    const float c = fv * to_f32<T>(c_prev[idx]) + iv * gv;
    i_s[idx] = iv;
    g_t[idx] = gv;
    f_s[idx] = fv;
    o_s[idx] = ov;
    c_out[idx] = from_f32<T>(c);
    h_out[idx] = from_f32<T>(ov * tanhf(c));
  }
}

std::vector<at::Tensor> lstm_gates_fwd(const at::Tensor& gates,
                                       const at::Tensor& c_prev,
                                       double forget_bias) {
  TORCH_CHECK(gates.is_cuda() && gates.dim() == 2 && gates.is_contiguous());
  const long B = gates.size(0);
  const long H = gates.size(1) / 4;
  TORCH_CHECK(c_prev.size(0) == B && c_prev.size(1) == H);
  auto cp = c_prev.contiguous();
  auto h_out = at::empty({B, H}, gates.options());
  auto c_out = at::empty({B, H}, gates.options());
  auto f32 = gates.options().dtype(at::kFloat);
  auto i_s = at::empty({B, H}, f32);
  auto g_t = at::empty({B, H}, f32);
  auto f_s = at::empty({B, H}, f32);
  auto o_s = at::empty({B, H}, f32);
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, gates.scalar_type(), "lstm_gates_fwd", [&] {
        lstm_gates_fwd_kernel<scalar_t>
            <<<px_grid(B * H), PX_BLOCK, 0, stream>>>(
                reinterpret_cast<const scalar_t*>(gates.data_ptr()),
                reinterpret_cast<const scalar_t*>(cp.data_ptr()),
                reinterpret_cast<scalar_t*>(h_out.data_ptr()),
                reinterpret_cast<scalar_t*>(c_out.data_ptr()),
                i_s.data_ptr<float>(), g_t.data_ptr<float>(),
                f_s.data_ptr<float>(), o_s.data_ptr<float>(), B * H, H,
                (float)forget_bias);
        HIP_CHECK_KERNEL();
      });
  return {h_out, c_out, i_s, g_t, f_s, o_s};
}

template <typename T>
__global__ void lstm_gates_bwd_kernel(
    const T* __restrict__ dh, const T* __restrict__ dc_in,
    const T* __restrict__ c_prev, const T* __restrict__ c_out,
    const float* __restrict__ i_s, const float* __restrict__ g_t,
    const float* __restrict__ f_s, const float* __restrict__ o_s,
    T* __restrict__ dgates, T* __restrict__ dc_prev, long bh, long hsz) {
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < bh;
       idx += (long)gridDim.x * blockDim.x) {
    const long b = idx / hsz;
    const long h = idx - b * hsz;
    const long g0 = b * 4 * hsz + h;
    const float dhv = to_f32<T>(dh[idx]);
    const float tc = tanhf(to_f32<T>(c_out[idx]));
    const float ov = o_s[idx];
    const float dov = dhv * tc;
    const float dc = dhv * ov * (1.f - tc * tc) + to_f32<T>(dc_in[idx]);
    const float iv = i_s[idx], gv = g_t[idx], fv = f_s[idx];
    const float div_ = dc * gv;
    const float dgv = dc * iv;
    const float dfv = dc * to_f32<T>(c_prev[idx]);
    dc_prev[idx] = from_f32<T>(dc * fv);
    dgates[g0] = from_f32<T>(div_ * iv * (1.f - iv));
    dgates[g0 + hsz] = from_f32<T>(dgv * (1.f - gv * gv));
    dgates[g0 + 2 * hsz] = from_f32<T>(dfv * fv * (1.f - fv));
    dgates[g0 + 3 * hsz] = from_f32<T>(dov * ov * (1.f - ov));
  }
}

std::vector<at::Tensor> lstm_gates_bwd(const at::Tensor& dh,
                                       const at::Tensor& dc_in,
                                       const at::Tensor& c_prev,
                                       const at::Tensor& c_out,
                                       const at::Tensor& i_s,
                                       const at::Tensor& g_t,
                                       const at::Tensor& f_s,
                                       const at::Tensor& o_s) {
  const long B = dh.size(0), H = dh.size(1);
  auto dgates = at::empty({B, 4 * H}, dh.options());
  auto dc_prev = at::empty({B, H}, dh.options());
  auto cp = c_prev.contiguous();
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, dh.scalar_type(), "lstm_gates_bwd", [&] {
        lstm_gates_bwd_kernel<scalar_t>
            <<<px_grid(B * H), PX_BLOCK, 0, stream>>>(
                reinterpret_cast<const scalar_t*>(dh.data_ptr()),
                reinterpret_cast<const scalar_t*>(dc_in.data_ptr()),
                reinterpret_cast<const scalar_t*>(cp.data_ptr()),
                reinterpret_cast<const scalar_t*>(c_out.data_ptr()),
                i_s.data_ptr<float>(), g_t.data_ptr<float>(),
                f_s.data_ptr<float>(), o_s.data_ptr<float>(),
                reinterpret_cast<scalar_t*>(dgates.data_ptr()),
                reinterpret_cast<scalar_t*>(dc_prev.data_ptr()), B * H, H);
        HIP_CHECK_KERNEL();
      });
  return {dgates, dc_prev};
}

// out-variants: write into caller-provided buffers (the sequence-level
// LSTM Function pre-allocates [T, B, H] stacks and passes per-step slices,
// so the time loop issues ZERO extra copy kernels).

void lstm_gates_fwd_out(const at::Tensor& gates, const at::Tensor& c_prev,
                        double forget_bias, at::Tensor h_out,
                        at::Tensor c_out, at::Tensor i_s, at::Tensor g_t,
                        at::Tensor f_s, at::Tensor o_s) {
  const long B = gates.size(0);
  const long H = gates.size(1) / 4;
  TORCH_CHECK(gates.is_contiguous() && c_prev.is_contiguous() &&
              h_out.is_contiguous() && c_out.is_contiguous());
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, gates.scalar_type(), "lstm_gates_fwd_out",
      [&] {
        lstm_gates_fwd_kernel<scalar_t>
            <<<px_grid(B * H), PX_BLOCK, 0, stream>>>(
                reinterpret_cast<const scalar_t*>(gates.data_ptr()),
                reinterpret_cast<const scalar_t*>(c_prev.data_ptr()),
                reinterpret_cast<scalar_t*>(h_out.data_ptr()),
                reinterpret_cast<scalar_t*>(c_out.data_ptr()),
                i_s.data_ptr<float>(), g_t.data_ptr<float>(),
                f_s.data_ptr<float>(), o_s.data_ptr<float>(), B * H, H,
                (float)forget_bias);
        HIP_CHECK_KERNEL();
      });
}

void lstm_gates_bwd_out(const at::Tensor& dh, const at::Tensor& dc_in,
                        const at::Tensor& c_prev, const at::Tensor& c_out,
                        const at::Tensor& i_s, const at::Tensor& g_t,
                        const at::Tensor& f_s, const at::Tensor& o_s,
                        at::Tensor dgates_out, at::Tensor dc_prev_out) {
  const long B = dh.size(0), H = dh.size(1);
  TORCH_CHECK(dh.is_contiguous() && dc_in.is_contiguous() &&
              dgates_out.is_contiguous() && dc_prev_out.is_contiguous());
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, dh.scalar_type(), "lstm_gates_bwd_out",
      [&] {
        lstm_gates_bwd_kernel<scalar_t>
            <<<px_grid(B * H), PX_BLOCK, 0, stream>>>(
                reinterpret_cast<const scalar_t*>(dh.data_ptr()),
                reinterpret_cast<const scalar_t*>(dc_in.data_ptr()),
                reinterpret_cast<const scalar_t*>(c_prev.data_ptr()),
                reinterpret_cast<const scalar_t*>(c_out.data_ptr()),
                i_s.data_ptr<float>(), g_t.data_ptr<float>(),
                f_s.data_ptr<float>(), o_s.data_ptr<float>(),
                reinterpret_cast<scalar_t*>(dgates_out.data_ptr()),
                reinterpret_cast<scalar_t*>(dc_prev_out.data_ptr()), B * H,
                H);
        HIP_CHECK_KERNEL();
      });
}
