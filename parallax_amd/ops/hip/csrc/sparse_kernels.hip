// Embedding-lookup gather + indexed-slices segmented row reduce (contract
// N2: the MI355X equivalent of the reference fork's
// SparseConditionalAccumulator take_indexed_slices_grad with
// average_option — SURVEY §2.3).
//
// rows_gather:      out[i, :] = weight[ids[i], :]       (HBM-bound gather)
// segment_sum_rows: out[s, :] = sum_k values[order[k], :] over the sorted
//                   run [seg_starts[s], seg_starts[s+1]); fp32 accumulation;
//                   optional per-row count average.
//
// Layout notes (CDNA4): lanes walk the row dimension contiguously so every
// wave issues coalesced 4-16 B/lane accesses; float rows of D%4==0 use
// float4 (16 B/lane — the coalescing sweet spot, guide G13).

#include <torch/extension.h>
#include <ATen/ATen.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

// ---------------------------------------------------------------------------
// rows_gather
// ---------------------------------------------------------------------------

template <typename T, typename VecT, int VEC>
__global__ void rows_gather_kernel(const T* __restrict__ w,
                                   const int64_t* __restrict__ ids,
                                   T* __restrict__ out, long n, long d,
                                   long nrows) {
  const long dv = d / VEC;
  const long total = n * dv;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long row = idx / dv;
    const long col = idx - row * dv;
    // clamp instead of faulting the GPU on an out-of-range id (torch's
    // index_select raises host-side; a device page fault here would take
    // the whole process down) — valid ids are unaffected
    long src = ids[row];
    src = src < 0 ? 0 : (src >= nrows ? nrows - 1 : src);
    reinterpret_cast<VecT*>(out)[row * dv + col] =
        reinterpret_cast<const VecT*>(w)[src * dv + col];
  }
}

at::Tensor rows_gather(const at::Tensor& weight, const at::Tensor& ids) {
  TORCH_CHECK(weight.is_cuda() && ids.is_cuda(), "rows_gather: GPU only");
  TORCH_CHECK(ids.scalar_type() == at::kLong);
  auto w = weight.contiguous();
  auto flat_ids = ids.contiguous().view(-1);
  const long n = flat_ids.numel();
  const long d = w.dim() == 1 ? 1 : w.size(1);
  std::vector<int64_t> out_shape(ids.sizes().begin(), ids.sizes().end());
  if (w.dim() > 1) out_shape.push_back(d);
  auto out = at::empty(out_shape, w.options());
  if (n == 0) return out;

  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, w.scalar_type(), "rows_gather", [&] {
        using T = scalar_t;
        const T* wp = reinterpret_cast<const T*>(w.data_ptr());
        T* op = reinterpret_cast<T*>(out.data_ptr());
        const int64_t* ip = flat_ids.data_ptr<int64_t>();
        auto stream = at::cuda::getCurrentCUDAStream();
        if (sizeof(T) == 4 && d % 4 == 0) {
          long total = n * (d / 4);
          rows_gather_kernel<T, float4, 4>
              <<<px_grid(total), PX_BLOCK, 0, stream>>>(wp, ip, op, n, d,
                                                        w.size(0));
        } else if (sizeof(T) == 2 && d % 8 == 0) {
          long total = n * (d / 8);
          rows_gather_kernel<T, float4, 8>
              <<<px_grid(total), PX_BLOCK, 0, stream>>>(wp, ip, op, n, d,
                                                        w.size(0));
        } else {
          long total = n * d;
          rows_gather_kernel<T, T, 1>
              <<<px_grid(total), PX_BLOCK, 0, stream>>>(wp, ip, op, n, d,
                                                        w.size(0));
        }
        HIP_CHECK_KERNEL();
      });
  return out;
}

// ---------------------------------------------------------------------------
// segment_sum_rows
// ---------------------------------------------------------------------------

// Scatter-atomic formulation: one thread per INPUT element (k, c) adds its
// value into the fp32 accumulator of its segment.  LOAD-BALANCED for any
// duplication profile — Zipfian candidate ids produce segments hundreds of
// entries long, which serialize a per-OUTPUT-element gather loop (measured
// 147 us vs ~20 us for this form at the LM1B softmax_w shape).  The
// segment of entry k is found by binary search over seg_starts (wave-
// uniform for 64 consecutive lanes of one k).  Atomic contention only
// occurs between duplicates of one id, spread across d columns.
template <typename T>
__global__ void segment_sum_scatter_kernel(
    const T* __restrict__ values, const int* __restrict__ order,
    const int* __restrict__ seg_starts, float* __restrict__ acc, long n,
    long u, long d) {
  const long total = n * d;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long k = idx / d;
    const long c = idx - k * d;
    // binary search: greatest s with seg_starts[s] <= k
    int lo = 0, hi = (int)u - 1;
    while (lo < hi) {
      int mid = (lo + hi + 1) >> 1;
      if (seg_starts[mid] <= (int)k) lo = mid; else hi = mid - 1;
    }
    const float v = to_f32<T>(values[(long)order[k] * d + c]);
    const int beg = seg_starts[lo];
    const int end = seg_starts[lo + 1];
    if (end - beg == 1) {
      acc[(long)lo * d + c] = v;  // unique id: plain store
    } else {
      atomicAdd(&acc[(long)lo * d + c], v);
    }
  }
}

template <typename T>
__global__ void segment_finalize_kernel(const float* __restrict__ acc,
                                        const int* __restrict__ seg_starts,
                                        T* __restrict__ out, long u, long d,
                                        bool average) {
  const long total = u * d;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long s = idx / d;
    float v = acc[idx];
    if (average) {
      v *= 1.0f / (float)(seg_starts[s + 1] - seg_starts[s]);
    }
    out[idx] = from_f32<T>(v);
  }
}

at::Tensor segment_sum_rows(const at::Tensor& values, const at::Tensor& order,
                            const at::Tensor& seg_starts, bool average) {
  TORCH_CHECK(values.is_cuda(), "segment_sum_rows: GPU only");
  TORCH_CHECK(order.scalar_type() == at::kInt &&
              seg_starts.scalar_type() == at::kInt);
  auto v = values.contiguous();
  auto ord = order.contiguous();
  auto seg = seg_starts.contiguous();
  const long u = seg.numel() - 1;
  const long d = v.dim() == 1 ? 1 : v.size(1);
  auto out = at::empty({u, d}, v.options());
  if (u == 0) return out;

  const long n = ord.numel();
  // fp32 accumulator (== out when out is already fp32 and no averaging
  // pass is needed separately)
  const bool direct = v.scalar_type() == at::kFloat;
  auto acc = direct ? out : at::empty({u, d}, v.options().dtype(at::kFloat));
  acc.zero_();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, v.scalar_type(), "segment_sum_rows", [&] {
        using T = scalar_t;
        auto stream = at::cuda::getCurrentCUDAStream();
        segment_sum_scatter_kernel<T>
            <<<px_grid(n * d, PX_BLOCK, 131072), PX_BLOCK, 0, stream>>>(
                reinterpret_cast<const T*>(v.data_ptr()),
                ord.data_ptr<int>(), seg.data_ptr<int>(),
                acc.data_ptr<float>(), n, u, d);
        HIP_CHECK_KERNEL();
        if (!direct || average) {
          segment_finalize_kernel<T>
              <<<px_grid(u * d, PX_BLOCK, 131072), PX_BLOCK, 0, stream>>>(
                  acc.data_ptr<float>(), seg.data_ptr<int>(),
                  reinterpret_cast<T*>(out.data_ptr()), u, d, average);
          HIP_CHECK_KERNEL();
        }
      });
  return out;
}
