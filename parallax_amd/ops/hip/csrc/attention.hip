// Fused scaled-dot-product attention for SHORT sequences (T, S <= 64,
// d_head = 64), bf16 in/out, fp32 math — the NMT transformer's shape
// (48 x 48 x 64 per head).
//
// Library flash-attention kernels are tiled for long sequences; at T=48
// they spend 17 us (fwd) / 28 us (bwd) on ~2 us of HBM traffic.  Here one
// 256-thread workgroup owns one (batch, head) pair and the ENTIRE
// attention computation lives in LDS: q/k/v (+dO in bwd) tiles are loaded
// once, scores/probabilities never touch global memory, and each output
// element is produced by exactly one thread (no atomics).
//
// Grid = B*H workgroups (512 for NMT) — an intentionally small launch:
// the kernel is tiny and latency-bound, not occupancy-bound.
//
// Backward recomputes P from q/k (cheaper than spilling it in fwd),
// then:  dV = P^T dO;  dP = dO V^T;  dS = P o (dP - rowsum(dP o P));
//        dQ = s * dS K;  dK = s * dS^T Q.
// P is kept in LDS as bf16 (halves the footprint; matches bf16 training
// precision), dP in fp32.

#include <torch/extension.h>
#include <ATen/ATen.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

#define ATT_MAX 64      // max rows (T) / cols (S) per tile
#define ATT_D 64        // head dim (fixed)
#define ATT_LD 68       // bf16 tile row stride (8B-aligned, off-bank)
#define ATT_LDF 65      // fp32 score row stride

typedef unsigned short ushort_t;

__device__ __forceinline__ float bf2f(ushort_t u) {
  union {
    unsigned int i;
    float f;
  } c;
  c.i = ((unsigned int)u) << 16;
  return c.f;
}

__device__ __forceinline__ ushort_t f2bf(float f) {
  union {
    unsigned int i;
    float f;
  } c;
  c.f = f;
  unsigned int lsb = (c.i >> 16) & 1;          // round-to-nearest-even
  return (ushort_t)((c.i + 0x7fff + lsb) >> 16);
}

// Load an [R, 64] bf16 tile (row stride rs elements in global) into LDS
// with row stride ATT_LD.  8-byte (4-elem) chunks, 256 threads.
__device__ __forceinline__ void att_load_tile(
    const ushort_t* __restrict__ g, long rs, int rows, ushort_t* lds) {
  const int chunks = rows * (ATT_D / 4);
  for (int i = threadIdx.x; i < chunks; i += blockDim.x) {
    const int r = i / (ATT_D / 4);
    const int c = (i % (ATT_D / 4)) * 4;
    *reinterpret_cast<uint2*>(&lds[r * ATT_LD + c]) =
        *reinterpret_cast<const uint2*>(&g[r * rs + c]);
  }
}

// dot of two 64-elem bf16 LDS rows (4-wide LDS reads)
__device__ __forceinline__ float att_dot64(const ushort_t* a,
                                           const ushort_t* b) {
  float acc = 0.f;
#pragma unroll
  for (int c = 0; c < ATT_D; c += 4) {
    uint2 av = *reinterpret_cast<const uint2*>(&a[c]);
    uint2 bv = *reinterpret_cast<const uint2*>(&b[c]);
    const ushort_t* ap = reinterpret_cast<const ushort_t*>(&av);
    const ushort_t* bp = reinterpret_cast<const ushort_t*>(&bv);
#pragma unroll
    for (int k = 0; k < 4; ++k) acc += bf2f(ap[k]) * bf2f(bp[k]);
  }
  return acc;
}

__global__ void __launch_bounds__(PX_BLOCK)
px_attn_fwd_kernel(const ushort_t* __restrict__ q,
                   const ushort_t* __restrict__ k,
                   const ushort_t* __restrict__ v,
                   ushort_t* __restrict__ o, int H, int T, int S,
                   float scale, int causal, long qb, long qh, long qt,
                   long kb, long kh, long kt, long vb, long vh, long vt) {
  __shared__ ushort_t lq[ATT_MAX * ATT_LD];
  __shared__ ushort_t lk[ATT_MAX * ATT_LD];
  __shared__ ushort_t lv[ATT_MAX * ATT_LD];
  __shared__ float lp[ATT_MAX * ATT_LDF];  // scores -> probs (fp32)
  const long b = blockIdx.x / H, h = blockIdx.x % H;
  att_load_tile(q + b * qb + h * qh, qt, T, lq);
  att_load_tile(k + b * kb + h * kh, kt, S, lk);
  att_load_tile(v + b * vb + h * vh, vt, S, lv);
  __syncthreads();
  // scores
  for (int i = threadIdx.x; i < T * S; i += blockDim.x) {
    const int t = i / S, s = i % S;
    lp[t * ATT_LDF + s] =
        (causal && s > t)
            ? -INFINITY
            : att_dot64(&lq[t * ATT_LD], &lk[s * ATT_LD]) * scale;
  }
  __syncthreads();
  // softmax: one thread per row
  for (int t = threadIdx.x; t < T; t += blockDim.x) {
    float* row = &lp[t * ATT_LDF];
    float mx = -INFINITY;
    for (int s = 0; s < S; ++s) mx = fmaxf(mx, row[s]);
    float sum = 0.f;
    for (int s = 0; s < S; ++s) {
      float e = __expf(row[s] - mx);
      row[s] = e;
      sum += e;
    }
    const float inv = 1.f / sum;
    for (int s = 0; s < S; ++s) row[s] *= inv;
  }
  __syncthreads();
  // O = P @ V  (o is a contiguous [B,H,T,64] allocation)
  ushort_t* og = o + (long)blockIdx.x * T * ATT_D;
  for (int i = threadIdx.x; i < T * ATT_D; i += blockDim.x) {
    const int t = i / ATT_D, d = i % ATT_D;
    const float* pr = &lp[t * ATT_LDF];
    float acc = 0.f;
    for (int s = 0; s < S; ++s) acc += pr[s] * bf2f(lv[s * ATT_LD + d]);
    og[t * ATT_D + d] = f2bf(acc);
  }
}

__global__ void __launch_bounds__(PX_BLOCK)
px_attn_bwd_kernel(const ushort_t* __restrict__ q,
                   const ushort_t* __restrict__ k,
                   const ushort_t* __restrict__ v,
                   const ushort_t* __restrict__ dout,
                   ushort_t* __restrict__ dq, ushort_t* __restrict__ dk,
                   ushort_t* __restrict__ dv, int H, int T, int S,
                   float scale, int causal, long qb, long qh, long qt,
                   long kb, long kh, long kt, long vb, long vh, long vt,
                   long db, long dh, long dt) {
  // q/k/v/dout carry their own (b, h, row) strides; dq/dk/dv are
  // contiguous [B,H,rows,64] allocations indexed directly.
  __shared__ ushort_t lq[ATT_MAX * ATT_LD];
  __shared__ ushort_t lk[ATT_MAX * ATT_LD];
  __shared__ ushort_t lv[ATT_MAX * ATT_LD];
  __shared__ ushort_t ldo[ATT_MAX * ATT_LD];
  __shared__ ushort_t lp[ATT_MAX * ATT_LD];   // P (bf16), then dS
  __shared__ float ldp[ATT_MAX * ATT_LDF];    // dP (fp32)
  __shared__ float ldelta[ATT_MAX];
  const long b = blockIdx.x / H, h = blockIdx.x % H;
  const long bh = blockIdx.x;
  att_load_tile(q + b * qb + h * qh, qt, T, lq);
  att_load_tile(k + b * kb + h * kh, kt, S, lk);
  att_load_tile(v + b * vb + h * vh, vt, S, lv);
  att_load_tile(dout + b * db + h * dh, dt, T, ldo);
  __syncthreads();
  // recompute scores into ldp (fp32 scratch)
  for (int i = threadIdx.x; i < T * S; i += blockDim.x) {
    const int t = i / S, s = i % S;
    ldp[t * ATT_LDF + s] =
        (causal && s > t)
            ? -INFINITY
            : att_dot64(&lq[t * ATT_LD], &lk[s * ATT_LD]) * scale;
  }
  __syncthreads();
  // softmax rows -> P (bf16 in lp)
  for (int t = threadIdx.x; t < T; t += blockDim.x) {
    float* row = &ldp[t * ATT_LDF];
    float mx = -INFINITY;
    for (int s = 0; s < S; ++s) mx = fmaxf(mx, row[s]);
    float sum = 0.f;
    for (int s = 0; s < S; ++s) {
      float e = __expf(row[s] - mx);
      row[s] = e;
      sum += e;
    }
    const float inv = 1.f / sum;
    for (int s = 0; s < S; ++s)
      lp[t * ATT_LD + s] = f2bf(row[s] * inv);
  }
  __syncthreads();
  // dV[s][d] = sum_t P[t][s] * dO[t][d]   (outputs are contiguous
  // [B,H,rows,64] allocations — fixed indexing)
  ushort_t* dvg = dv + bh * (long)S * ATT_D;
  for (int i = threadIdx.x; i < S * ATT_D; i += blockDim.x) {
    const int s = i / ATT_D, d = i % ATT_D;
    float acc = 0.f;
    for (int t = 0; t < T; ++t)
      acc += bf2f(lp[t * ATT_LD + s]) * bf2f(ldo[t * ATT_LD + d]);
    dvg[s * ATT_D + d] = f2bf(acc);
  }
  // dP[t][s] = dO[t] . V[s]
  for (int i = threadIdx.x; i < T * S; i += blockDim.x) {
    const int t = i / S, s = i % S;
    ldp[t * ATT_LDF + s] =
        att_dot64(&ldo[t * ATT_LD], &lv[s * ATT_LD]);
  }
  __syncthreads();
  // delta[t] = sum_s dP*P ; then dS (overwrites lp, scaled)
  for (int t = threadIdx.x; t < T; t += blockDim.x) {
    float acc = 0.f;
    for (int s = 0; s < S; ++s)
      acc += ldp[t * ATT_LDF + s] * bf2f(lp[t * ATT_LD + s]);
    ldelta[t] = acc;
  }
  __syncthreads();
  for (int i = threadIdx.x; i < T * S; i += blockDim.x) {
    const int t = i / S, s = i % S;
    const float p = bf2f(lp[t * ATT_LD + s]);
    lp[t * ATT_LD + s] =
        f2bf(p * (ldp[t * ATT_LDF + s] - ldelta[t]) * scale);
  }
  __syncthreads();
  // dQ[t][d] = sum_s dS[t][s] * K[s][d]
  ushort_t* dqg = dq + bh * (long)T * ATT_D;
  for (int i = threadIdx.x; i < T * ATT_D; i += blockDim.x) {
    const int t = i / ATT_D, d = i % ATT_D;
    float acc = 0.f;
    for (int s = 0; s < S; ++s)
      acc += bf2f(lp[t * ATT_LD + s]) * bf2f(lk[s * ATT_LD + d]);
    dqg[t * ATT_D + d] = f2bf(acc);
  }
  // dK[s][d] = sum_t dS[t][s] * Q[t][d]
  ushort_t* dkg = dk + bh * (long)S * ATT_D;
  for (int i = threadIdx.x; i < S * ATT_D; i += blockDim.x) {
    const int s = i / ATT_D, d = i % ATT_D;
    float acc = 0.f;
    for (int t = 0; t < T; ++t)
      acc += bf2f(lp[t * ATT_LD + s]) * bf2f(lq[t * ATT_LD + d]);
    dkg[s * ATT_D + d] = f2bf(acc);
  }
}

static void att_check(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == at::kBFloat16 &&
                  t.dim() == 4 && t.stride(3) == 1 && t.size(3) == ATT_D,
              "px_attn: ", name,
              " must be CUDA bf16 [B,H,T,64] with contiguous last dim");
  // 8-byte global loads: every row must start 4-element aligned
  TORCH_CHECK(t.stride(0) % 4 == 0 && t.stride(1) % 4 == 0 &&
                  t.stride(2) % 4 == 0,
              "px_attn: ", name, " strides must be multiples of 4");
}

at::Tensor px_attn_fwd(const at::Tensor& q, const at::Tensor& k,
                       const at::Tensor& v, bool causal) {
  att_check(q, "q");
  att_check(k, "k");
  att_check(v, "v");
  const long B = q.size(0), H = q.size(1), T = q.size(2), S = k.size(2);
  TORCH_CHECK(T <= ATT_MAX && S <= ATT_MAX, "px_attn: T,S must be <= 64");
  TORCH_CHECK(k.size(0) == B && k.size(1) == H && v.sizes() == k.sizes());
  auto o = at::empty({B, H, T, ATT_D}, q.options());
  const float scale = 1.0f / sqrtf((float)ATT_D);
  auto stream = at::cuda::getCurrentCUDAStream();
  px_attn_fwd_kernel<<<(int)(B * H), PX_BLOCK, 0, stream>>>(
      reinterpret_cast<const ushort_t*>(q.data_ptr()),
      reinterpret_cast<const ushort_t*>(k.data_ptr()),
      reinterpret_cast<const ushort_t*>(v.data_ptr()),
      reinterpret_cast<ushort_t*>(o.data_ptr()), (int)H, (int)T, (int)S,
      scale, causal ? 1 : 0, q.stride(0), q.stride(1), q.stride(2),
      k.stride(0), k.stride(1), k.stride(2), v.stride(0), v.stride(1),
      v.stride(2));
  HIP_CHECK_KERNEL();
  return o;
}

std::vector<at::Tensor> px_attn_bwd(const at::Tensor& q,
                                    const at::Tensor& k,
                                    const at::Tensor& v,
                                    const at::Tensor& dout, bool causal) {
  att_check(q, "q");
  att_check(k, "k");
  att_check(v, "v");
  att_check(dout, "dout");
  const long B = q.size(0), H = q.size(1), T = q.size(2), S = k.size(2);
  TORCH_CHECK(T <= ATT_MAX && S <= ATT_MAX, "px_attn: T,S must be <= 64");
  auto dc = dout.contiguous();
  auto dq = at::empty({B, H, T, ATT_D}, q.options());
  auto dk = at::empty({B, H, S, ATT_D}, q.options());
  auto dv = at::empty({B, H, S, ATT_D}, q.options());
  const float scale = 1.0f / sqrtf((float)ATT_D);
  auto stream = at::cuda::getCurrentCUDAStream();
  px_attn_bwd_kernel<<<(int)(B * H), PX_BLOCK, 0, stream>>>(
      reinterpret_cast<const ushort_t*>(q.data_ptr()),
      reinterpret_cast<const ushort_t*>(k.data_ptr()),
      reinterpret_cast<const ushort_t*>(v.data_ptr()),
      reinterpret_cast<const ushort_t*>(dc.data_ptr()),
      reinterpret_cast<ushort_t*>(dq.data_ptr()),
      reinterpret_cast<ushort_t*>(dk.data_ptr()),
      reinterpret_cast<ushort_t*>(dv.data_ptr()), (int)H, (int)T, (int)S,
      scale, causal ? 1 : 0, q.stride(0), q.stride(1), q.stride(2),
      k.stride(0), k.stride(1), k.stride(2), v.stride(0), v.stride(1),
      v.stride(2), dc.stride(0), dc.stride(1), dc.stride(2));
  HIP_CHECK_KERNEL();
  return {dq, dk, dv};
}
