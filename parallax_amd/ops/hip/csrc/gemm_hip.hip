#include "hip/hip_runtime.h"
// Hand-written CDNA4 MFMA GEMM: C[M,N] = A[M,K] @ B[N,K]^T, bf16 in /
// bf16 out, fp32 accumulate.  The building block for the NHWC conv path
// (a 1x1 convolution IS this GEMM with A = activations [N*H*W, Cin] and
// B = weights [Cout, Cin]) and for framework-owned projection GEMMs.
//
// Structure (guide cdna_hip_programming.md §5, the LDS-staged ladder):
//   * 128x128 output tile per 256-thread block (4 waves, each a 64x64
//     sub-tile of 4x4 mfma_f32_16x16x32_bf16 fragments, 64 fp32 acc
//     VGPRs/lane);
//   * BK=64 K-step, both operand tiles staged to LDS by
//     global_load_lds_dwordx4 (16 B/lane direct-to-LDS DMA), double
//     buffered;
//   * XOR swizzle byte^=((row&7)<<4) against ds_read_b128 bank conflicts
//     (guide G4/T2) applied on the glds SOURCE address + the LDS read
//     (rule 21: the destination stays lane-linear);
//   * 2-phase schedule: issue next tile's DMA, compute current from LDS
//     behind s_setprio(1) (T3 minimum form + T5);
//   * XCD-aware bijective block swizzle (T1) for L2 locality.
// Both operands are K-major (B passed TRANSPOSED, i.e. [N,K] row-major,
// the natural layout of torch Linear/conv weights), so every MFMA
// fragment is 8 contiguous bf16 per lane.

#include <torch/extension.h>
#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define BM 128
#define BN 128
#define BK 64
#define PXG_THREADS 256

// LDS tile: [128 rows][64 k] bf16 = 16 KiB; byte row stride 128.
// swizzle: byte ^= ((row & 7) << 4)   (16-byte granularity, involution)
__device__ __forceinline__ int swz(int row, int kbyte) {
  return row * (BK * 2) + (kbyte ^ ((row & 7) << 4));
}

// Cadd: optional bf16 addend with arbitrary row stride (elements),
// fused into the epilogue (the addmm beta=1 case: gates = gx[:,t] + h@W).
__global__ __launch_bounds__(PXG_THREADS, 2)
void pxgemm_bt_kernel(const ushort* __restrict__ A,
                      const ushort* __restrict__ B,
                      ushort* __restrict__ C, long M, long N, long K,
                      int nbm, int nbn, const ushort* __restrict__ Cadd,
                      long add_rstride) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // [2 buffers][A|B][16 KiB]
#define LDSA(b) (smem + (b) * 32768)
#define LDSB(b) (smem + (b) * 32768 + 16384)

  // XCD-aware bijective remap of the flat block id (guide T1)
  int bid = blockIdx.x;
  {
    const int nwg = nbm * nbn;
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = bid % 8, pos = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const int bm = (bid / nbn) * BM;
  const int bn = (bid % nbn) * BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  // wave grid 2x2: each wave owns a 64x64 sub-tile
  const int wm = (wave >> 1) * 64;
  const int wn = (wave & 1) * 64;
  const int fr = lane & 15;        // fragment row/col within 16
  const int ko = (lane >> 4) * 8;  // fragment k offset (8 bf16)

  f32x4 acc[4][4] = {};

  // ---- glds staging: each thread DMAs 16 B per instruction; a 16 KiB
  // tile needs 4 instructions.  LDS dest is lane-linear
  // (base + flat_byte), source address carries the inverse swizzle.
  const long a_row0 = bm, b_row0 = bn;
  // flat element position for this thread's i-th piece:
  //   byte p = (tid + i*256) * 16 ; row = p / 128 ; kbyte = p % 128
  int p_row[4], p_kb[4];
  long a_src_row[4], b_src_row[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int p = (tid + i * PXG_THREADS) * 16;
    p_row[i] = p / (BK * 2);
    p_kb[i] = (p % (BK * 2)) ^ ((p_row[i] & 7) << 4);  // inverse swizzle
    // edge tiles: clamp out-of-range source rows to the last valid row —
    // the stale fragments they produce are discarded by the guarded
    // epilogue stores, so NO host-side padding copies are needed
    long ar = a_row0 + p_row[i];
    long br = b_row0 + p_row[i];
    a_src_row[i] = ar < M ? ar : M - 1;
    b_src_row[i] = br < N ? br : N - 1;
  }

  const int ktiles = (int)(K / BK);

#define STAGE(buf, kt)                                                    \
  do {                                                                     \
    const long kbase = (long)(kt) * BK;                                    \
    _Pragma("unroll") for (int i = 0; i < 4; ++i) {                        \
      const ushort* srcA = A + a_src_row[i] * K + kbase +                  \
                           (p_kb[i] >> 1);                                 \
      const ushort* srcB = B + b_src_row[i] * K + kbase +                  \
                           (p_kb[i] >> 1);                                 \
      __builtin_amdgcn_global_load_lds(                                    \
          (const __attribute__((address_space(1))) void*)srcA,             \
          (__attribute__((address_space(3))) void*)(LDSA(buf) +            \
                                                    (tid + i * 256) * 16), \
          16, 0, 0);                                                       \
      __builtin_amdgcn_global_load_lds(                                    \
          (const __attribute__((address_space(1))) void*)srcB,             \
          (__attribute__((address_space(3))) void*)(LDSB(buf) +            \
                                                    (tid + i * 256) * 16), \
          16, 0, 0);                                                       \
    }                                                                      \
  } while (0)

  STAGE(0, 0);
  __builtin_amdgcn_s_waitcnt(/*vmcnt(0) lgkmcnt(0)*/ 0);
  __builtin_amdgcn_s_barrier();

  int cur = 0;
  for (int kt = 0; kt < ktiles; ++kt) {
    if (kt + 1 < ktiles) {
      STAGE(cur ^ 1, kt + 1);
    }
    // compute current tile from LDS
#pragma unroll
    for (int kk = 0; kk < BK / 32; ++kk) {
      bf16x8 a_frag[4], b_frag[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int ar = wm + i * 16 + fr;
        a_frag[i] = *(const bf16x8*)(LDSA(cur) +
                                     swz(ar, (kk * 32 + ko) * 2));
        const int br = wn + i * 16 + fr;
        b_frag[i] = *(const bf16x8*)(LDSB(cur) +
                                     swz(br, (kk * 32 + ko) * 2));
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 4; ++i) {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[i], b_frag[j], acc[i][j], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
    // drain this iteration's DMA and cross the barrier
    __builtin_amdgcn_s_waitcnt(0);
    __builtin_amdgcn_s_barrier();
    cur ^= 1;
  }

  // ---- epilogue: C/D fragment map for 16x16x32: col = lane&15,
  // row = (lane>>4)*4 + reg (guide §3); bf16 convert + store
  const int crow0 = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long row = bm + wm + i * 16 + crow0 + r;
        const long col = bn + wn + j * 16 + fr;
        if (row >= M || col >= N) continue;
        float fv = acc[i][j][r];
        if (Cadd != nullptr) {
          __hip_bfloat16 av =
              *reinterpret_cast<const __hip_bfloat16*>(
                  &Cadd[row * add_rstride + col]);
          fv += __bfloat162float(av);
        }
        __hip_bfloat16 v = __float2bfloat16(fv);
        C[row * N + col] = *reinterpret_cast<ushort*>(&v);
      }
    }
  }
}

at::Tensor pxgemm_bt_add(const at::Tensor& a, const at::Tensor& bt,
                         const c10::optional<at::Tensor>& addend) {
  TORCH_CHECK(a.is_cuda() && bt.is_cuda(), "pxgemm_bt: GPU only");
  TORCH_CHECK(a.scalar_type() == at::kBFloat16 &&
              bt.scalar_type() == at::kBFloat16, "pxgemm_bt: bf16 only");
  TORCH_CHECK(a.dim() == 2 && bt.dim() == 2 && a.is_contiguous() &&
              bt.is_contiguous());
  const long M = a.size(0), K = a.size(1), N = bt.size(0);
  TORCH_CHECK(bt.size(1) == K, "pxgemm_bt: K mismatch");
  TORCH_CHECK(K % BK == 0, "pxgemm_bt needs K % 64 == 0");
  TORCH_CHECK(M >= 1 && N >= 16, "pxgemm_bt: degenerate shape");
  auto c = at::empty({M, N}, a.options());
  const int nbm = (int)((M + BM - 1) / BM);
  const int nbn = (int)((N + BN - 1) / BN);
  const int smem = 64 * 1024;
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  hipFuncSetAttribute(
      reinterpret_cast<const void*>(&pxgemm_bt_kernel),
      hipFuncAttributeMaxDynamicSharedMemorySize, smem);
  const ushort* add_ptr = nullptr;
  long add_rs = 0;
  if (addend.has_value()) {
    const auto& ad = addend.value();
    TORCH_CHECK(ad.scalar_type() == at::kBFloat16 && ad.dim() == 2 &&
                ad.size(0) == M && ad.size(1) == N && ad.stride(1) == 1,
                "pxgemm_bt: bad addend");
    add_ptr = reinterpret_cast<const ushort*>(ad.data_ptr());
    add_rs = ad.stride(0);
  }
 hipLaunchKernelGGL(( pxgemm_bt_kernel), dim3(nbm * nbn), dim3(PXG_THREADS), smem, stream, 
      reinterpret_cast<const ushort*>(a.data_ptr()),
      reinterpret_cast<const ushort*>(bt.data_ptr()),
      reinterpret_cast<ushort*>(c.data_ptr()), M, N, K, nbm, nbn,
      add_ptr, add_rs);
  HIP_CHECK_KERNEL();
  return c;
}

at::Tensor pxgemm_bt(const at::Tensor& a, const at::Tensor& bt) {
  return pxgemm_bt_add(a, bt, c10::nullopt);
}

// ---------------------------------------------------------------------------
// Implicit-GEMM 3x3 NHWC convolution forward (stride 1 or 2, padding 1)
// on the same 128x128 LDS-staged MFMA structure: the A operand is the
// im2col view of the HOST-PADDED input (glds cannot predicate, so the
// halo is materialized once per call), the B operand is the weight
// reshaped [Cout, 9*Cin] k-major (its natural channels_last order).
// out[(n,oy,ox), co] = sum_{dy,dx,ci} xp[n][oy*s+dy][ox*s+dx][ci]
//                                     * w[co][dy][dx][ci]
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(PXG_THREADS, 2)
void pxconv3x3_kernel(const ushort* __restrict__ XP,  // [N,H+2,W+2,Cin]
                      const ushort* __restrict__ B,   // [Cout, 9*Cin]
                      ushort* __restrict__ C,         // [M, Cout]
                      long M, long N, long K, int nbm, int nbn,
                      int OH, int OW, int WP /*W+2*/, int HP /*H+2*/,
                      int Cin, int log2cin, int stride) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  int bid = blockIdx.x;
  {
    const int nwg = nbm * nbn;
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = bid % 8, pos = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const int bm = (bid / nbn) * BM;
  const int bn = (bid % nbn) * BN;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = (wave >> 1) * 64;
  const int wn = (wave & 1) * 64;
  const int fr = lane & 15;
  const int ko = (lane >> 4) * 8;
  f32x4 acc[4][4] = {};

  int p_row[4], p_kb[4];
  long a_base[4], b_src_row[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int p = (tid + i * PXG_THREADS) * 16;
    p_row[i] = p / (BK * 2);
    p_kb[i] = (p % (BK * 2)) ^ ((p_row[i] & 7) << 4);
    long ar = bm + p_row[i];
    if (ar >= M) ar = M - 1;          // clamped edge tile
    long br = bn + p_row[i];
    b_src_row[i] = br < N ? br : N - 1;
    // im2col row -> (n, oy, ox) -> padded-input byte base (k-independent)
    const long n_img = ar / ((long)OH * OW);
    const long rem = ar - n_img * ((long)OH * OW);
    const int oy = (int)(rem / OW);
    const int ox = (int)(rem - (long)oy * OW);
    a_base[i] = (((n_img * HP) + (long)oy * stride) * WP +
                 (long)ox * stride) * Cin;
  }
  const int ktiles = (int)(K / BK);

#define CSTAGE(buf, kt)                                                    \
  do {                                                                     \
    const int kbase = (kt) * BK;                                           \
    _Pragma("unroll") for (int i = 0; i < 4; ++i) {                        \
      const int ka = kbase + (p_kb[i] >> 1);                               \
      const int dydx = ka >> log2cin;                                      \
      const int ci = ka - (dydx << log2cin);                               \
      const int dy = dydx >= 6 ? 2 : (dydx >= 3 ? 1 : 0);                  \
      const int dx = dydx - 3 * dy;                                        \
      const ushort* srcA =                                                 \
          XP + a_base[i] + ((long)dy * WP + dx) * Cin + ci;                \
      const ushort* srcB = B + b_src_row[i] * K + kbase +                  \
                           (p_kb[i] >> 1);                                 \
      __builtin_amdgcn_global_load_lds(                                    \
          (const __attribute__((address_space(1))) void*)srcA,             \
          (__attribute__((address_space(3))) void*)(LDSA(buf) +            \
                                                    (tid + i * 256) * 16), \
          16, 0, 0);                                                       \
      __builtin_amdgcn_global_load_lds(                                    \
          (const __attribute__((address_space(1))) void*)srcB,             \
          (__attribute__((address_space(3))) void*)(LDSB(buf) +            \
                                                    (tid + i * 256) * 16), \
          16, 0, 0);                                                       \
    }                                                                      \
  } while (0)

  CSTAGE(0, 0);
  __builtin_amdgcn_s_waitcnt(0);
  __builtin_amdgcn_s_barrier();
  int cur = 0;
  for (int kt = 0; kt < ktiles; ++kt) {
    if (kt + 1 < ktiles) {
      CSTAGE(cur ^ 1, kt + 1);
    }
#pragma unroll
    for (int kk = 0; kk < BK / 32; ++kk) {
      bf16x8 a_frag[4], b_frag[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int ar = wm + i * 16 + fr;
        a_frag[i] = *(const bf16x8*)(LDSA(cur) +
                                     swz(ar, (kk * 32 + ko) * 2));
        const int br = wn + i * 16 + fr;
        b_frag[i] = *(const bf16x8*)(LDSB(cur) +
                                     swz(br, (kk * 32 + ko) * 2));
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 4; ++i) {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[i], b_frag[j], acc[i][j], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
    __builtin_amdgcn_s_waitcnt(0);
    __builtin_amdgcn_s_barrier();
    cur ^= 1;
  }
  const int crow0 = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long row = bm + wm + i * 16 + crow0 + r;
        const long col = bn + wn + j * 16 + fr;
        if (row >= M || col >= N) continue;
        __hip_bfloat16 v = __float2bfloat16(acc[i][j][r]);
        C[row * N + col] = *reinterpret_cast<ushort*>(&v);
      }
    }
  }
}

at::Tensor pxconv3x3_fwd(const at::Tensor& xp, const at::Tensor& w2d,
                         long n_img, long oh, long ow, long cin,
                         long stride, long HP, long WP) {
  // xp: padded NHWC activations flattened [N*(H+2)*(W+2)*Cin] bf16 view;
  // w2d: [Cout, 9*Cin] bf16 (channels_last weight order)
  TORCH_CHECK(xp.is_cuda() && w2d.is_cuda());
  TORCH_CHECK(xp.scalar_type() == at::kBFloat16 &&
              w2d.scalar_type() == at::kBFloat16);
  TORCH_CHECK(xp.is_contiguous() && w2d.is_contiguous());
  TORCH_CHECK((cin & (cin - 1)) == 0 && cin >= 64,
              "pxconv3x3: Cin must be a power of two >= 64");
  const long cout = w2d.size(0);
  const long K = w2d.size(1);
  TORCH_CHECK(K == 9 * cin && K % BK == 0);
  const long M = n_img * oh * ow;
  // the kernel touches y_in <= (oh-1)*stride+2 < HP and likewise for x
  TORCH_CHECK((oh - 1) * stride + 3 <= HP && (ow - 1) * stride + 3 <= WP,
              "pxconv3x3: padded dims too small");
  TORCH_CHECK(xp.numel() == n_img * HP * WP * cin,
              "pxconv3x3: padded input shape mismatch");
  int log2cin = 0;
  while ((1 << log2cin) < cin) log2cin++;
  auto c = at::empty({M, cout}, xp.options());
  const int nbm = (int)((M + BM - 1) / BM);
  const int nbn = (int)((cout + BN - 1) / BN);
  const int smem = 64 * 1024;
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  hipFuncSetAttribute(
      reinterpret_cast<const void*>(&pxconv3x3_kernel),
      hipFuncAttributeMaxDynamicSharedMemorySize, smem);
 hipLaunchKernelGGL(( pxconv3x3_kernel), dim3(nbm * nbn), dim3(PXG_THREADS), smem, stream, 
      reinterpret_cast<const ushort*>(xp.data_ptr()),
      reinterpret_cast<const ushort*>(w2d.data_ptr()),
      reinterpret_cast<ushort*>(c.data_ptr()), M, cout, K, nbm, nbn,
      (int)oh, (int)ow, (int)WP, (int)HP, (int)cin, log2cin, (int)stride);
  HIP_CHECK_KERNEL();
  return c;
}
