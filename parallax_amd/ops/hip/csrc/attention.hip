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
// P/dS are kept in LDS as bf16; dP lives entirely in registers.
// All matrix phases are register-tiled (see att_scores_softmax): each
// wave owns a block of rows, each lane one column, so every LDS element
// is read once per wave instead of once per MAC.

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

// ---------------------------------------------------------------------------
// Register-tiled phases.  The naive per-(t,s) dot re-reads BOTH operand
// rows from LDS for every MAC (~0.6 MB/WG) and is LDS-bandwidth-bound;
// here each wave owns a contiguous block of rows and each LANE owns one
// column (s or d), so every LDS element is read ONCE per wave (operands
// amortized in registers) and row reductions happen across lanes.
#define ATT_RPW (ATT_MAX / 4)  // max rows per wave (4 waves / block)
#define ATT_CHUNK 16           // d-chunk cached in registers

// scores + softmax -> P (bf16) in lp.  Wave w owns rows
// [w*rpw, w*rpw+rows); lane owns column s = LANE.
__device__ __forceinline__ void att_scores_softmax(
    const ushort_t* lq, const ushort_t* lk, ushort_t* lp, int T, int S,
    float scale, int causal) {
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int rpw = (T + 3) / 4;
  const int t0 = wid * rpw;
  const int rows = min(rpw, T - t0) > 0 ? min(rpw, T - t0) : 0;
  float acc[ATT_RPW];
#pragma unroll
  for (int r = 0; r < ATT_RPW; ++r) acc[r] = 0.f;
  // k[s=lane] fragment per d-chunk, reused across the wave's rows
  for (int c0 = 0; c0 < ATT_D; c0 += ATT_CHUNK) {
    float kf[ATT_CHUNK];
#pragma unroll
    for (int j = 0; j < ATT_CHUNK; ++j)
      kf[j] = (lane < S) ? bf2f(lk[lane * ATT_LD + c0 + j]) : 0.f;
#pragma unroll
    for (int r = 0; r < ATT_RPW; ++r) {
      if (r >= rows) break;
      const ushort_t* qr = &lq[(t0 + r) * ATT_LD + c0];
      float a = 0.f;
#pragma unroll
      for (int j = 0; j < ATT_CHUNK; ++j) a += bf2f(qr[j]) * kf[j];
      acc[r] += a;
    }
  }
  // per-row softmax across lanes (lane == s)
#pragma unroll
  for (int r = 0; r < ATT_RPW; ++r) {
    if (r >= rows) break;
    const int t = t0 + r;
    float sc = (lane < S && !(causal && lane > t))
                   ? acc[r] * scale
                   : -INFINITY;
    float mx = wave_reduce_max(sc);
    mx = __shfl(mx, 0, WAVE);
    float e = (sc == -INFINITY) ? 0.f : __expf(sc - mx);
    float sum = wave_reduce_sum(e);
    sum = __shfl(sum, 0, WAVE);
    if (lane < S) lp[t * ATT_LD + lane] = f2bf(e / sum);
  }
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
  __shared__ ushort_t lp[ATT_MAX * ATT_LD];  // probabilities (bf16)
  const long b = blockIdx.x / H, h = blockIdx.x % H;
  att_load_tile(q + b * qb + h * qh, qt, T, lq);
  att_load_tile(k + b * kb + h * kh, kt, S, lk);
  att_load_tile(v + b * vb + h * vh, vt, S, lv);
  __syncthreads();
  att_scores_softmax(lq, lk, lp, T, S, scale, causal);
  __syncthreads();
  // O = P @ V: lane owns column d; each v element read once per wave
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int rpw = (T + 3) / 4;
  const int t0 = wid * rpw;
  const int rows = min(rpw, T - t0) > 0 ? min(rpw, T - t0) : 0;
  float oacc[ATT_RPW];
#pragma unroll
  for (int r = 0; r < ATT_RPW; ++r) oacc[r] = 0.f;
  for (int s = 0; s < S; ++s) {
    const float vv = bf2f(lv[s * ATT_LD + lane]);
#pragma unroll
    for (int r = 0; r < ATT_RPW; ++r) {
      if (r >= rows) break;
      oacc[r] += bf2f(lp[(t0 + r) * ATT_LD + s]) * vv;
    }
  }
  ushort_t* og = o + (long)blockIdx.x * T * ATT_D;
#pragma unroll
  for (int r = 0; r < ATT_RPW; ++r)
    if (r < rows) og[(t0 + r) * ATT_D + lane] = f2bf(oacc[r]);
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
  __shared__ ushort_t lp[ATT_MAX * ATT_LD];  // P (bf16), then dS
  const long b = blockIdx.x / H, h = blockIdx.x % H;
  const long bh = blockIdx.x;
  att_load_tile(q + b * qb + h * qh, qt, T, lq);
  att_load_tile(k + b * kb + h * kh, kt, S, lk);
  att_load_tile(v + b * vb + h * vh, vt, S, lv);
  att_load_tile(dout + b * db + h * dh, dt, T, ldo);
  __syncthreads();
  att_scores_softmax(lq, lk, lp, T, S, scale, causal);
  __syncthreads();
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int rpw = (T + 3) / 4;       // q-side rows per wave
  const int spw = (S + 3) / 4;       // k/v-side rows per wave
  const int t0 = wid * rpw;
  const int trows = min(rpw, T - t0) > 0 ? min(rpw, T - t0) : 0;
  const int s0 = wid * spw;
  const int srows = min(spw, S - s0) > 0 ? min(spw, S - s0) : 0;
  // dV[s][d] = sum_t P[t][s] dO[t][d]: lane owns d, wave owns s rows
  {
    float acc[ATT_RPW];
#pragma unroll
    for (int r = 0; r < ATT_RPW; ++r) acc[r] = 0.f;
    for (int t = 0; t < T; ++t) {
      const float dov = bf2f(ldo[t * ATT_LD + lane]);
#pragma unroll
      for (int r = 0; r < ATT_RPW; ++r) {
        if (r >= srows) break;
        acc[r] += bf2f(lp[t * ATT_LD + (s0 + r)]) * dov;
      }
    }
    ushort_t* dvg = dv + bh * (long)S * ATT_D;
#pragma unroll
    for (int r = 0; r < ATT_RPW; ++r)
      if (r < srows) dvg[(s0 + r) * ATT_D + lane] = f2bf(acc[r]);
  }
  __syncthreads();  // dV read EVERY P row; dS below overwrites lp
  // dP[t][s] = dO[t] . V[s]; delta[t] = sum_s dP*P; dS overwrites lp.
  // lane owns s; wave owns t rows (same mapping as the score phase)
  {
    float acc[ATT_RPW];
#pragma unroll
    for (int r = 0; r < ATT_RPW; ++r) acc[r] = 0.f;
    for (int c0 = 0; c0 < ATT_D; c0 += ATT_CHUNK) {
      float vf[ATT_CHUNK];
#pragma unroll
      for (int j = 0; j < ATT_CHUNK; ++j)
        vf[j] = (lane < S) ? bf2f(lv[lane * ATT_LD + c0 + j]) : 0.f;
#pragma unroll
      for (int r = 0; r < ATT_RPW; ++r) {
        if (r >= trows) break;
        const ushort_t* dor = &ldo[(t0 + r) * ATT_LD + c0];
        float a = 0.f;
#pragma unroll
        for (int j = 0; j < ATT_CHUNK; ++j) a += bf2f(dor[j]) * vf[j];
        acc[r] += a;
      }
    }
#pragma unroll
    for (int r = 0; r < ATT_RPW; ++r) {
      if (r >= trows) break;
      const int t = t0 + r;
      const float pv = (lane < S) ? bf2f(lp[t * ATT_LD + lane]) : 0.f;
      float delta = wave_reduce_sum(pv * acc[r]);
      delta = __shfl(delta, 0, WAVE);
      if (lane < S)
        lp[t * ATT_LD + lane] = f2bf(pv * (acc[r] - delta) * scale);
    }
  }
  __syncthreads();
  // dQ[t][d] = sum_s dS[t][s] K[s][d]: lane owns d, wave owns t rows
  {
    float acc[ATT_RPW];
#pragma unroll
    for (int r = 0; r < ATT_RPW; ++r) acc[r] = 0.f;
    for (int s = 0; s < S; ++s) {
      const float kv = bf2f(lk[s * ATT_LD + lane]);
#pragma unroll
      for (int r = 0; r < ATT_RPW; ++r) {
        if (r >= trows) break;
        acc[r] += bf2f(lp[(t0 + r) * ATT_LD + s]) * kv;
      }
    }
    ushort_t* dqg = dq + bh * (long)T * ATT_D;
#pragma unroll
    for (int r = 0; r < ATT_RPW; ++r)
      if (r < trows) dqg[(t0 + r) * ATT_D + lane] = f2bf(acc[r]);
  }
  // dK[s][d] = sum_t dS[t][s] Q[t][d]: lane owns d, wave owns s rows
  {
    float acc[ATT_RPW];
#pragma unroll
    for (int r = 0; r < ATT_RPW; ++r) acc[r] = 0.f;
    for (int t = 0; t < T; ++t) {
      const float qv = bf2f(lq[t * ATT_LD + lane]);
#pragma unroll
      for (int r = 0; r < ATT_RPW; ++r) {
        if (r >= srows) break;
        acc[r] += bf2f(lp[t * ATT_LD + (s0 + r)]) * qv;
      }
    }
    ushort_t* dkg = dk + bh * (long)S * ATT_D;
#pragma unroll
    for (int r = 0; r < ATT_RPW; ++r)
      if (r < srows) dkg[(s0 + r) * ATT_D + lane] = f2bf(acc[r]);
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
