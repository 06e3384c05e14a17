// Common device helpers for the parallax_amd CDNA4 (gfx950) kernels.
//
// Conventions (per the CDNA HIP programming guide):
//   * wavefront = 64 lanes (hard-coded WAVE, never 32);
//   * blocks are multiples of 64 (default 256 = 4 waves);
//   * memory-bound kernels grid-stride with grid <= MAX_GRID blocks;
//   * reductions: wave shuffle first, then one LDS slot per wave;
//   * fp32 accumulation regardless of storage dtype.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64
#define PX_BLOCK 256
#define PX_MAX_GRID 2048

#define HIP_CHECK_KERNEL()                                            \
  do {                                                                 \
    hipError_t e = hipGetLastError();                                  \
    TORCH_CHECK(e == hipSuccess, "HIP kernel launch failed: ",         \
                hipGetErrorString(e));                                 \
  } while (0)

__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    v += __shfl_down(v, off, WAVE);
  }
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    v = fmaxf(v, __shfl_down(v, off, WAVE));
  }
  return v;
}

// Block reduce (sum) for blockDim.x == PX_BLOCK (4 waves).  Returns the
// reduced value on every thread.  `scratch` must hold >= 4 floats.
__device__ __forceinline__ float block_reduce_sum(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  const int nwaves = blockDim.x / WAVE;
  float r = (threadIdx.x < nwaves) ? scratch[threadIdx.x] : 0.0f;
  r = wave_reduce_sum(r);  // only wave 0's low lanes meaningful
  if (threadIdx.x == 0) scratch[0] = r;
  __syncthreads();
  float out = scratch[0];
  __syncthreads();
  return out;
}

__device__ __forceinline__ float block_reduce_max(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_reduce_max(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  const int nwaves = blockDim.x / WAVE;
  float r = (threadIdx.x < nwaves) ? scratch[threadIdx.x] : -INFINITY;
  r = wave_reduce_max(r);
  if (threadIdx.x == 0) scratch[0] = r;
  __syncthreads();
  float out = scratch[0];
  __syncthreads();
  return out;
}

// dtype conversion helpers (work for float/double and the c10 half types,
// whose host/device conversion operators c10 provides) ----------------------
template <typename T>
__device__ __forceinline__ float to_f32(T v) {
  return static_cast<float>(v);
}

template <typename T>
__device__ __forceinline__ T from_f32(float v) {
  return static_cast<T>(v);
}

// 4-wide vector of T with natural vector alignment: dwordx4 loads for
// fp32, dwordx2 for bf16/fp16 — one wave instruction moves 1 KB / 512 B
// instead of 256 B / 128 B of scalar traffic.
template <typename T>
struct alignas(4 * sizeof(T)) PxV4 {
  T v[4];
};

template <typename T>
__device__ __forceinline__ PxV4<T> px_ld4(const T* p, long j) {
  return *reinterpret_cast<const PxV4<T>*>(p + j);
}

template <typename T>
__device__ __forceinline__ void px_st4(T* p, long j, const PxV4<T>& x) {
  *reinterpret_cast<PxV4<T>*>(p + j) = x;
}

static inline int px_grid(long total, int block = PX_BLOCK,
                          int max_grid = PX_MAX_GRID) {
  long g = (total + block - 1) / block;
  if (g > max_grid) g = max_grid;
  if (g < 1) g = 1;
  return (int)g;
}
