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


// ===========================================================================
// MFMA variants (round 2): the short-seq shape on the matrix cores.
// One workgroup = 4 waves = one (b, h); every GEMM is
// mfma_f32_16x16x32_bf16 over zero-padded 64x64 LDS tiles; fragments are
// read as X[row][k-contiguous] / Y'[col][k-contiguous], so every
// reduction over a NON-contiguous axis gets a transposed twin image
// written once at load (or after the producing stage).  Row-wise softmax
// reductions are 4-step __shfl_xor within each 16-lane group (the C
// fragment holds col = lane&15, row = (lane>>4)*4 + reg).
// ===========================================================================

typedef __attribute__((ext_vector_type(8))) short att_bf16x8;
typedef __attribute__((ext_vector_type(4))) float att_f32x4;

// load a 16-byte (8-elem) A/B fragment from an LDS tile with row stride
// ATT_LD (136 B: 8-aligned, not 16) as two b64 reads
__device__ __forceinline__ att_bf16x8 att_frag(const ushort_t* lds,
                                               int row, int k0) {
  const ushort_t* p = lds + row * ATT_LD + k0;
  att_bf16x8 f;
  *reinterpret_cast<uint2*>(&f) = *reinterpret_cast<const uint2*>(p);
  *(reinterpret_cast<uint2*>(&f) + 1) =
      *reinterpret_cast<const uint2*>(p + 4);
  return f;
}

// load an [R,64] global tile into lds ([64][ATT_LD], rows >= R zeroed)
// and, when ldsT != nullptr, its transpose into ldsT[64][ATT_LD]
__device__ __forceinline__ void att_load_tile_t(
    const ushort_t* __restrict__ g, long rs, int rows, ushort_t* lds,
    ushort_t* ldsT) {
  for (int i = threadIdx.x; i < 64 * (ATT_D / 4); i += blockDim.x) {
    const int r = i / (ATT_D / 4);
    const int c = (i % (ATT_D / 4)) * 4;
    uint2 v;
    if (r < rows) {
      v = *reinterpret_cast<const uint2*>(&g[r * rs + c]);
    } else {
      v.x = 0u;
      v.y = 0u;
    }
    *reinterpret_cast<uint2*>(&lds[r * ATT_LD + c]) = v;
    if (ldsT != nullptr) {
      const ushort_t* e = reinterpret_cast<const ushort_t*>(&v);
#pragma unroll
      for (int j = 0; j < 4; ++j) ldsT[(c + j) * ATT_LD + r] = e[j];
    }
  }
}

// row-wise reduce over the 16 lanes of a C-fragment lane group
__device__ __forceinline__ float att_gmax(float v) {
#pragma unroll
  for (int w = 1; w < 16; w <<= 1) v = fmaxf(v, __shfl_xor(v, w, 64));
  return v;
}
__device__ __forceinline__ float att_gsum(float v) {
#pragma unroll
  for (int w = 1; w < 16; w <<= 1) v += __shfl_xor(v, w, 64);
  return v;
}

#define ATT_TILE_B (64 * ATT_LD)  // elems per LDS tile

// compute one 16x64 score row-block: acc[j] += A_rows(w16+fr) @ B_rows
__device__ __forceinline__ void att_qkt(const ushort_t* A,
                                        const ushort_t* B, int arow0,
                                        att_f32x4* acc) {
  const int lane = threadIdx.x & 63;
  const int fr = lane & 15;
  const int ko = (lane >> 4) * 8;
#pragma unroll
  for (int kk = 0; kk < 2; ++kk) {
    const att_bf16x8 a = att_frag(A, arow0 + fr, kk * 32 + ko);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const att_bf16x8 b = att_frag(B, j * 16 + fr, kk * 32 + ko);
      acc[j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[j], 0,
                                                       0, 0);
    }
  }
}

// write a C-layout fragment block (rows w16+crow0+r, cols j*16+cl) of
// bf16 values into an LDS tile, optionally also its transpose
__device__ __forceinline__ void att_cwrite(ushort_t* lds, ushort_t* ldsT,
                                           int row0, const float* vals4,
                                           int j) {
  const int lane = threadIdx.x & 63;
  const int cl = lane & 15;
  const int crow0 = (lane >> 4) * 4;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const ushort_t bv = f2bf(vals4[r]);
    const int row = row0 + crow0 + r;
    const int col = j * 16 + cl;
    lds[row * ATT_LD + col] = bv;
    if (ldsT != nullptr) ldsT[col * ATT_LD + row] = bv;
  }
}

__global__ __launch_bounds__(256)
void px_attn_fwd_mfma_kernel(
    const ushort_t* __restrict__ q, const ushort_t* __restrict__ k,
    const ushort_t* __restrict__ v, ushort_t* __restrict__ out, int B,
    int H, int T, int S, float scale, int causal, long qb, long qh,
    long qt, long kb, long kh, long kt, long vb, long vh, long vt,
    long ob_, long oh_, long ot_) {
  __shared__ ushort_t lq[ATT_TILE_B];
  __shared__ ushort_t lk[ATT_TILE_B];
  __shared__ ushort_t lvT[ATT_TILE_B];
  __shared__ ushort_t lp[ATT_TILE_B];
  const int bh = blockIdx.x;
  const int b = bh / H, h = bh % H;
  att_load_tile_t(q + b * qb + h * qh, qt, T, lq, nullptr);
  // K plain for QK^T; V transposed only: load V rows into lvT transpose
  for (int i = threadIdx.x; i < 64 * (ATT_D / 4); i += blockDim.x) {
    const int r = i / (ATT_D / 4);
    const int c = (i % (ATT_D / 4)) * 4;
    uint2 kv, vv;
    if (r < S) {
      kv = *reinterpret_cast<const uint2*>(&k[b * kb + h * kh + r * kt
                                              + c]);
      vv = *reinterpret_cast<const uint2*>(&v[b * vb + h * vh + r * vt
                                              + c]);
    } else {
      kv.x = kv.y = 0u;
      vv.x = vv.y = 0u;
    }
    *reinterpret_cast<uint2*>(&lk[r * ATT_LD + c]) = kv;
    const ushort_t* e = reinterpret_cast<const ushort_t*>(&vv);
#pragma unroll
    for (int j = 0; j < 4; ++j) lvT[(c + j) * ATT_LD + r] = e[j];
  }
  __syncthreads();

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int cl = lane & 15;
  const int crow0 = (lane >> 4) * 4;
  const int row0 = wave * 16;

  att_f32x4 sc[4] = {};
  att_qkt(lq, lk, row0, sc);

  // mask + softmax (rows row0+crow0+r, cols j*16+cl)
  float p[4][4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = row0 + crow0 + r;
    float mx = -1e30f;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float sv = sc[j][r] * scale;
      const int col = j * 16 + cl;
      if (col >= S || (causal && col > row)) sv = -1e30f;
      p[j][r] = sv;
      mx = fmaxf(mx, sv);
    }
    mx = att_gmax(mx);
    float sum = 0.f;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float e = p[j][r] > -1e29f ? __expf(p[j][r] - mx) : 0.f;
      p[j][r] = e;
      sum += e;
    }
    sum = att_gsum(sum);
    const float inv = sum > 0.f ? 1.f / sum : 0.f;
#pragma unroll
    for (int j = 0; j < 4; ++j) p[j][r] *= inv;
  }
  // P -> LDS (plain layout only)
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    float col4[4] = {p[j][0], p[j][1], p[j][2], p[j][3]};
    att_cwrite(lp, nullptr, row0, col4, j);
  }
  __syncthreads();

  // O = P @ V  (B' = V^T image)
  att_f32x4 oc[4] = {};
  att_qkt(lp, lvT, row0, oc);

  // strided store (the host allocates out [B,T,H,D]-permuted so the
  // model's transpose(1,2).reshape afterwards is a free view)
  ushort_t* ob = out + b * ob_ + h * oh_;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = row0 + crow0 + r;
      if (row < T) ob[row * ot_ + j * 16 + cl] = f2bf(oc[j][r]);
    }
  }
}

__global__ __launch_bounds__(256)
void px_attn_bwd_mfma_kernel(
    const ushort_t* __restrict__ q, const ushort_t* __restrict__ k,
    const ushort_t* __restrict__ v, const ushort_t* __restrict__ dout,
    ushort_t* __restrict__ dq, ushort_t* __restrict__ dk,
    ushort_t* __restrict__ dv, int B, int H, int T, int S, float scale,
    int causal, long qb, long qh, long qt, long kb, long kh, long kt,
    long vb, long vh, long vt, long db, long dh, long dt) {
  extern __shared__ __attribute__((aligned(16))) char attsm[];
  ushort_t* lq = reinterpret_cast<ushort_t*>(attsm);
  ushort_t* lqT = lq + ATT_TILE_B;
  ushort_t* lk = lqT + ATT_TILE_B;
  ushort_t* lkT = lk + ATT_TILE_B;
  ushort_t* lv = lkT + ATT_TILE_B;
  ushort_t* ldo = lv + ATT_TILE_B;
  ushort_t* ldoT = ldo + ATT_TILE_B;
  ushort_t* lpT = ldoT + ATT_TILE_B;
  // dS images ALIAS lv/ldo (both dead once dP is computed; a barrier
  // separates the last dP read from the first dS write) — 8 tiles of
  // LDS instead of 10 keeps 2 workgroups per CU resident
  ushort_t* lds_ = ldo;                  // dS plain
  ushort_t* ldsT = lv;                   // dS transposed
  const int bh = blockIdx.x;
  const int b = bh / H, h = bh % H;
  att_load_tile_t(q + b * qb + h * qh, qt, T, lq, lqT);
  att_load_tile_t(k + b * kb + h * kh, kt, S, lk, lkT);
  att_load_tile_t(v + b * vb + h * vh, vt, S, lv, nullptr);
  att_load_tile_t(dout + b * db + h * dh, dt, T, ldo, ldoT);
  __syncthreads();

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int cl = lane & 15;
  const int crow0 = (lane >> 4) * 4;
  const int row0 = wave * 16;

  // recompute P (rows = t) and dP, both in C-fragment layout
  att_f32x4 sc[4] = {};
  att_qkt(lq, lk, row0, sc);
  float p[4][4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = row0 + crow0 + r;
    float mx = -1e30f;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float sv = sc[j][r] * scale;
      const int col = j * 16 + cl;
      if (col >= S || (causal && col > row)) sv = -1e30f;
      p[j][r] = sv;
      mx = fmaxf(mx, sv);
    }
    mx = att_gmax(mx);
    float sum = 0.f;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float e = p[j][r] > -1e29f ? __expf(p[j][r] - mx) : 0.f;
      p[j][r] = e;
      sum += e;
    }
    sum = att_gsum(sum);
    const float inv = sum > 0.f ? 1.f / sum : 0.f;
#pragma unroll
    for (int j = 0; j < 4; ++j) p[j][r] *= inv;
  }
  // write P^T (needed by dV); P itself stays in registers for dS
#pragma unroll
  for (int j = 0; j < 4; ++j) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = row0 + crow0 + r;
      const int col = j * 16 + cl;
      lpT[col * ATT_LD + row] = f2bf(p[j][r]);
    }
  }
  // dP = dO @ V^T   (A' = dO, B' = V, both plain k(d)-major)
  att_f32x4 dpc[4] = {};
  att_qkt(ldo, lv, row0, dpc);
  __syncthreads();  // every wave done reading lv/ldo before dS aliases
  // dS = P o (dP - rowsum(dP o P)) * scale
  float dsv[4][4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    float rs = 0.f;
#pragma unroll
    for (int j = 0; j < 4; ++j) rs += dpc[j][r] * p[j][r];
    rs = att_gsum(rs);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      dsv[j][r] = p[j][r] * (dpc[j][r] - rs) * scale;
    }
  }
#pragma unroll
  for (int j = 0; j < 4; ++j) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = row0 + crow0 + r;
      const int col = j * 16 + cl;
      const ushort_t bv = f2bf(dsv[j][r]);
      lds_[row * ATT_LD + col] = bv;
      ldsT[col * ATT_LD + row] = bv;
    }
  }
  __syncthreads();

  // dQ = dS @ K        (A' = dS,  B' = K^T)
  att_f32x4 dqc[4] = {};
  att_qkt(lds_, lkT, row0, dqc);
  // dK = dS^T @ Q      (A' = dS^T, B' = Q^T)
  att_f32x4 dkc[4] = {};
  att_qkt(ldsT, lqT, row0, dkc);
  // dV = P^T @ dO      (A' = P^T,  B' = dO^T)
  att_f32x4 dvc[4] = {};
  att_qkt(lpT, ldoT, row0, dvc);

  // grads are [B,T|S,H,64] contiguous ([B,H,*,64] permuted views on the
  // host): element (b,h,row,col) at ((b*rows + row)*H + h)*64 + col
  const long grt = (long)H * ATT_D;
  ushort_t* dqb = dq + ((long)b * T) * grt + (long)h * ATT_D;
  ushort_t* dkb = dk + ((long)b * S) * grt + (long)h * ATT_D;
  ushort_t* dvb = dv + ((long)b * S) * grt + (long)h * ATT_D;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = row0 + crow0 + r;
      const int col = j * 16 + cl;
      if (row < T) dqb[row * grt + col] = f2bf(dqc[j][r]);
      if (row < S) {
        dkb[row * grt + col] = f2bf(dkc[j][r]);
        dvb[row * grt + col] = f2bf(dvc[j][r]);
      }
    }
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
  static const bool use_mfma = [] {
    const char* e = getenv("PARALLAX_ATTN_MFMA");
    return e == nullptr || e[0] != '0';
  }();
  if (use_mfma) {
    // allocate [B,T,H,D] and return the [B,H,T,D] permutation: the
    // model's transpose(1,2).reshape(B,T,H*D) is then a free view
    auto ob = at::empty({B, T, H, ATT_D}, q.options());
    auto ov = ob.permute({0, 2, 1, 3});
    px_attn_fwd_mfma_kernel<<<(int)(B * H), 256, 0, stream>>>(
        reinterpret_cast<const ushort_t*>(q.data_ptr()),
        reinterpret_cast<const ushort_t*>(k.data_ptr()),
        reinterpret_cast<const ushort_t*>(v.data_ptr()),
        reinterpret_cast<ushort_t*>(ob.data_ptr()), (int)B, (int)H,
        (int)T, (int)S, scale, causal ? 1 : 0, q.stride(0), q.stride(1),
        q.stride(2), k.stride(0), k.stride(1), k.stride(2), v.stride(0),
        v.stride(1), v.stride(2), ov.stride(0), ov.stride(1),
        ov.stride(2));
    HIP_CHECK_KERNEL();
    return ov;
  }
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
  const bool d_ok = dout.stride(3) == 1 && dout.stride(0) % 4 == 0 &&
                    dout.stride(1) % 4 == 0 && dout.stride(2) % 4 == 0;
  auto dc = d_ok ? dout : dout.contiguous();
  // grads as [B,T,H,D] permutations: the qkv-chunk cat in the caller's
  // backward then reads them contiguously
  auto dqb_ = at::empty({B, T, H, ATT_D}, q.options());
  auto dkb_ = at::empty({B, S, H, ATT_D}, q.options());
  auto dvb_ = at::empty({B, S, H, ATT_D}, q.options());
  auto dq = dqb_.permute({0, 2, 1, 3});
  auto dk = dkb_.permute({0, 2, 1, 3});
  auto dv = dvb_.permute({0, 2, 1, 3});
  const float scale = 1.0f / sqrtf((float)ATT_D);
  auto stream = at::cuda::getCurrentCUDAStream();
  static const bool use_mfma = [] {
    const char* e = getenv("PARALLAX_ATTN_MFMA");
    return e == nullptr || e[0] != '0';
  }();
  if (use_mfma) {
    const int smem = 8 * ATT_TILE_B * 2;
    hipFuncSetAttribute(
        reinterpret_cast<const void*>(&px_attn_bwd_mfma_kernel),
        hipFuncAttributeMaxDynamicSharedMemorySize, smem);
    px_attn_bwd_mfma_kernel<<<(int)(B * H), 256, smem, stream>>>(
        reinterpret_cast<const ushort_t*>(q.data_ptr()),
        reinterpret_cast<const ushort_t*>(k.data_ptr()),
        reinterpret_cast<const ushort_t*>(v.data_ptr()),
        reinterpret_cast<const ushort_t*>(dc.data_ptr()),
        reinterpret_cast<ushort_t*>(dqb_.data_ptr()),
        reinterpret_cast<ushort_t*>(dkb_.data_ptr()),
        reinterpret_cast<ushort_t*>(dvb_.data_ptr()), (int)B, (int)H,
        (int)T, (int)S, scale, causal ? 1 : 0, q.stride(0), q.stride(1),
        q.stride(2), k.stride(0), k.stride(1), k.stride(2), v.stride(0),
        v.stride(1), v.stride(2), dc.stride(0), dc.stride(1),
        dc.stride(2));
    HIP_CHECK_KERNEL();
    return {dq, dk, dv};
  }
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
