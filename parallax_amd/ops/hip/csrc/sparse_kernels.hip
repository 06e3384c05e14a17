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
                                   T* __restrict__ out, long n, long d) {
  const long dv = d / VEC;
  const long total = n * dv;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long row = idx / dv;
    const long col = idx - row * dv;
    const long src = ids[row];
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
              <<<px_grid(total), PX_BLOCK, 0, stream>>>(wp, ip, op, n, d);
        } else if (sizeof(T) == 2 && d % 8 == 0) {
          long total = n * (d / 8);
          rows_gather_kernel<T, float4, 8>
              <<<px_grid(total), PX_BLOCK, 0, stream>>>(wp, ip, op, n, d);
        } else {
          long total = n * d;
          rows_gather_kernel<T, T, 1>
              <<<px_grid(total), PX_BLOCK, 0, stream>>>(wp, ip, op, n, d);
        }
        HIP_CHECK_KERNEL();
      });
  return out;
}

// ---------------------------------------------------------------------------
// segment_sum_rows
// ---------------------------------------------------------------------------

// Flat (segment, column) indexing: one thread per output element with the
// per-segment gather loop inside.  Segments from a deduplicated gradient
// are short (few duplicates per id), so the inner loop is 1-3 iterations;
// flat indexing keeps all 64 lanes busy for ANY d (the block-per-segment
// form wastes 255/256 threads on d=1 bias tables).
template <typename T>
__global__ void segment_sum_rows_kernel(const T* __restrict__ values,
                                        const int* __restrict__ order,
                                        const int* __restrict__ seg_starts,
                                        T* __restrict__ out, long u, long d,
                                        bool average) {
  const long total = u * d;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long s = idx / d;
    const long c = idx - s * d;
    const int beg = seg_starts[s];
    const int end = seg_starts[s + 1];
    float acc = 0.0f;
    for (int k = beg; k < end; ++k) {
      acc += to_f32<T>(values[(long)order[k] * d + c]);
    }
    if (average) acc *= 1.0f / (float)(end - beg);
    out[idx] = from_f32<T>(acc);
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

  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, v.scalar_type(), "segment_sum_rows", [&] {
        using T = scalar_t;
        auto stream = at::cuda::getCurrentCUDAStream();
        // one element per thread: the per-element chain is two DEPENDENT
        // loads (order[k] -> values row), so latency must be hidden by
        // thread-level parallelism — do not cap the grid at PX_MAX_GRID
        int grid = px_grid(u * d, PX_BLOCK, 131072);
        segment_sum_rows_kernel<T>
            <<<grid, PX_BLOCK, 0, stream>>>(
                reinterpret_cast<const T*>(v.data_ptr()),
                ord.data_ptr<int>(), seg.data_ptr<int>(),
                reinterpret_cast<T*>(out.data_ptr()), u, d, average);
        HIP_CHECK_KERNEL();
      });
  return out;
}
