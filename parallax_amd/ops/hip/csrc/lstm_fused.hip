// Fused LSTM recurrent step: the per-timestep gates GEMM and the gate
// pointwise math in ONE kernel launch (round-2 of the LM1B hot loop;
// reference cell: examples/lm1b/language_model.py:51-62).
//
//   fwd : gates = hp_{t-1} @ W_h^T + gx_t   (M=B, N=4H, K=P)
//         i,g,f,o = act(gates); c_t = f*c_{t-1} + i*g; s_t = o*tanh(c_t)
//   bwd : ds = dHP_t @ w_proj^T             (M=B, N=H, K=P)
//         dgates_t, dc_{t-1} = gate_bwd(ds, dc_t, acts)
//
// Why fuse: at B=128 these GEMMs are LATENCY-bound (guide
// cdna_hip_programming.md §5 "M=256 sampling/decode projection GEMM"):
// per-step cost is one dependent-chain launch + the [B,4H] gates tensor's
// HBM round trip + a second pointwise launch.  Grouping each block's
// columns as (i|g|f|o) slices of the SAME 64 hidden units makes the
// pointwise REGISTER-LOCAL in the MFMA C fragments (all four gates of a
// hidden unit land in the same lane), so the gates tensor never exists in
// memory and the chain is one kernel.
//
// Structure per block (BB batch rows x BH hidden units):
//   * waves = BB/16, each owns 16 rows; acc = 4 gates x BH/16 col-frags;
//   * K-loop BK=64, A (hp) and B (W rows of the owned columns) staged to
//     LDS by global_load_lds_dwordx4, double-buffered, XOR-swizzled
//     (byte ^= (row&7)<<4) against ds_read_b128 bank conflicts (T2/G4);
//   * grid = (H/BH) x (B/BB) blocks — sized so the per-step W re-read
//     (8 MB, L2-resident across timesteps) streams in parallel.

#include <torch/extension.h>
#include <ATen/ATen.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define LBK 64

__device__ __forceinline__ float px_sigmoidf(float x) {
  return 1.f / (1.f + __expf(-x));
}

// LDS tile addressing: row stride 128 B (LBK*2), 16-B XOR swizzle
__device__ __forceinline__ int lswz(int row, int kbyte) {
  return row * (LBK * 2) + (kbyte ^ ((row & 7) << 4));
}

// ---------------------------------------------------------------------------
// forward: one timestep, gates GEMM + pointwise
// ---------------------------------------------------------------------------
// BB: batch rows per block (multiple of 16); BH: hidden units per block.
// A = hp [B, P] bf16 (contiguous); Bmat = whT [4H, P] bf16 (w_h
// transposed once per sequence); gx strided [B, 4H] slice (row stride
// gx_rs elements).
template <int BB, int BH>
__global__ __launch_bounds__(BB * 4, 2)
void lstm_step_fwd_kernel(const ushort* __restrict__ hp,
                          const ushort* __restrict__ whT,
                          const ushort* __restrict__ gx, long gx_rs,
                          const ushort* __restrict__ c_prev,
                          ushort* __restrict__ s_out,
                          ushort* __restrict__ c_out,
                          float* __restrict__ i_s, float* __restrict__ g_t,
                          float* __restrict__ f_s, float* __restrict__ o_s,
                          int Bsz, int H, int P, float forget_bias) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  constexpr int THREADS = BB * 4;
  constexpr int A_BYTES = BB * LBK * 2;        // per K-tile
  constexpr int B_ROWS = 4 * BH;               // gate-grouped columns
  constexpr int B_BYTES = B_ROWS * LBK * 2;
  constexpr int PIECES_A = A_BYTES / (THREADS * 16);
  constexpr int PIECES_B = B_BYTES / (THREADS * 16);
  constexpr int NF = BH / 16;                  // col frags per gate
#define LFA(b) (smem + (b) * (A_BYTES + B_BYTES))
#define LFB(b) (smem + (b) * (A_BYTES + B_BYTES) + A_BYTES)

  const int nbh = H / BH;
  const int j0 = (blockIdx.x % nbh) * BH;      // hidden-unit base
  const int row0 = (blockIdx.x / nbh) * BB;    // batch-row base
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int fr = lane & 15;
  const int ko = (lane >> 4) * 8;

  f32x4 acc[4][NF] = {};

  // glds piece addressing (16 B per thread per piece, lane-linear dest)
  int pa_row[PIECES_A], pa_kb[PIECES_A];
  long a_src_row[PIECES_A];
#pragma unroll
  for (int i = 0; i < PIECES_A; ++i) {
    const int p = (tid + i * THREADS) * 16;
    pa_row[i] = p / (LBK * 2);
    pa_kb[i] = (p % (LBK * 2)) ^ ((pa_row[i] & 7) << 4);
    long ar = row0 + pa_row[i];
    a_src_row[i] = ar < Bsz ? ar : Bsz - 1;
  }
  int pb_row[PIECES_B], pb_kb[PIECES_B];
  long b_src_row[PIECES_B];
#pragma unroll
  for (int i = 0; i < PIECES_B; ++i) {
    const int p = (tid + i * THREADS) * 16;
    pb_row[i] = p / (LBK * 2);
    pb_kb[i] = (p % (LBK * 2)) ^ ((pb_row[i] & 7) << 4);
    const int gate = pb_row[i] / BH;
    const int col = pb_row[i] - gate * BH;
    b_src_row[i] = (long)gate * H + j0 + col;   // row in whT [4H, P]
  }

  const int ktiles = P / LBK;

#define LSTAGE(buf, kt)                                                   \
  do {                                                                     \
    const long kbase = (long)(kt) * LBK;                                   \
    _Pragma("unroll") for (int i = 0; i < PIECES_A; ++i) {                 \
      const ushort* src = hp + a_src_row[i] * P + kbase + (pa_kb[i] >> 1); \
      __builtin_amdgcn_global_load_lds(                                    \
          (const __attribute__((address_space(1))) void*)src,              \
          (__attribute__((address_space(3))) void*)(LFA(buf) +             \
              (tid + i * THREADS) * 16), 16, 0, 0);                        \
    }                                                                      \
    _Pragma("unroll") for (int i = 0; i < PIECES_B; ++i) {                 \
      const ushort* src = whT + b_src_row[i] * P + kbase +                 \
                          (pb_kb[i] >> 1);                                 \
      __builtin_amdgcn_global_load_lds(                                    \
          (const __attribute__((address_space(1))) void*)src,              \
          (__attribute__((address_space(3))) void*)(LFB(buf) +             \
              (tid + i * THREADS) * 16), 16, 0, 0);                        \
    }                                                                      \
  } while (0)

  LSTAGE(0, 0);
  __builtin_amdgcn_s_waitcnt(0);
  __builtin_amdgcn_s_barrier();

  int cur = 0;
  for (int kt = 0; kt < ktiles; ++kt) {
    if (kt + 1 < ktiles) LSTAGE(cur ^ 1, kt + 1);
#pragma unroll
    for (int kk = 0; kk < LBK / 32; ++kk) {
      bf16x8 a_frag;
      {
        const int ar = wave * 16 + fr;
        a_frag = *(const bf16x8*)(LFA(cur) + lswz(ar, (kk * 32 + ko) * 2));
      }
      bf16x8 b_frag[4][NF];
#pragma unroll
      for (int g = 0; g < 4; ++g) {
#pragma unroll
        for (int nf = 0; nf < NF; ++nf) {
          const int br = g * BH + nf * 16 + fr;
          b_frag[g][nf] =
              *(const bf16x8*)(LFB(cur) + lswz(br, (kk * 32 + ko) * 2));
        }
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int g = 0; g < 4; ++g) {
#pragma unroll
        for (int nf = 0; nf < NF; ++nf) {
          acc[g][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag, b_frag[g][nf], acc[g][nf], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
    __builtin_amdgcn_s_waitcnt(0);
    __builtin_amdgcn_s_barrier();
    cur ^= 1;
  }

  // epilogue: add gx, run the gate math in registers, store outputs.
  // C/D map (16x16x32): col = lane&15, row = (lane>>4)*4 + r.
  const int crow0 = (lane >> 4) * 4;
#pragma unroll
  for (int nf = 0; nf < NF; ++nf) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const long row = row0 + wave * 16 + crow0 + r;
      if (row >= Bsz) continue;
      const long col = j0 + nf * 16 + fr;           // hidden unit index
      const long gbase = row * gx_rs + col;
      const float gi = acc[0][nf][r] +
          to_f32<__hip_bfloat16>(((const __hip_bfloat16*)gx)[gbase]);
      const float gg = acc[1][nf][r] +
          to_f32<__hip_bfloat16>(((const __hip_bfloat16*)gx)[gbase + H]);
      const float gf = acc[2][nf][r] +
          to_f32<__hip_bfloat16>(
              ((const __hip_bfloat16*)gx)[gbase + 2 * H]);
      const float go = acc[3][nf][r] +
          to_f32<__hip_bfloat16>(
              ((const __hip_bfloat16*)gx)[gbase + 3 * H]);
      const float iv = px_sigmoidf(gi);
      const float gv = tanhf(gg);
      const float fv = px_sigmoidf(gf + forget_bias);
      const float ov = px_sigmoidf(go);
      const long hidx = row * H + col;
      const float c = fv * to_f32<__hip_bfloat16>(
                               ((const __hip_bfloat16*)c_prev)[hidx]) +
                      iv * gv;
      i_s[hidx] = iv;
      g_t[hidx] = gv;
      f_s[hidx] = fv;
      o_s[hidx] = ov;
      __hip_bfloat16 cb = __float2bfloat16(c);
      c_out[hidx] = *reinterpret_cast<ushort*>(&cb);
      __hip_bfloat16 sb = __float2bfloat16(ov * tanhf(c));
      s_out[hidx] = *reinterpret_cast<ushort*>(&sb);
    }
  }
#undef LFA
#undef LFB
#undef LSTAGE
}

void lstm_step_fwd_fused(const at::Tensor& hp, const at::Tensor& whT,
                         const at::Tensor& gx_t, const at::Tensor& c_prev,
                         double forget_bias, at::Tensor s_out,
                         at::Tensor c_out, at::Tensor i_s, at::Tensor g_t,
                         at::Tensor f_s, at::Tensor o_s) {
  TORCH_CHECK(hp.is_cuda() && hp.scalar_type() == at::kBFloat16 &&
              whT.scalar_type() == at::kBFloat16 &&
              gx_t.scalar_type() == at::kBFloat16);
  TORCH_CHECK(hp.is_contiguous() && whT.is_contiguous() &&
              c_prev.is_contiguous() && s_out.is_contiguous() &&
              c_out.is_contiguous());
  TORCH_CHECK(gx_t.stride(1) == 1, "gx slice must be row-strided only");
  const int B = (int)hp.size(0), P = (int)hp.size(1);
  const int H = (int)c_prev.size(1);
  TORCH_CHECK(whT.size(0) == 4 * (long)H && whT.size(1) == P);
  TORCH_CHECK(P % LBK == 0 && H % 32 == 0 && B <= 128);
  constexpr int BB = 64, BH = 32;
  TORCH_CHECK(B % BB == 0, "lstm fused fwd: B must be a multiple of 64");
  const int blocks = (H / BH) * (B / BB);
  const int smem = 2 * (BB * LBK * 2 + 4 * BH * LBK * 2);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipFuncSetAttribute(
      reinterpret_cast<const void*>(&lstm_step_fwd_kernel<BB, BH>),
      hipFuncAttributeMaxDynamicSharedMemorySize, smem);
  lstm_step_fwd_kernel<BB, BH><<<blocks, BB * 4, smem, stream>>>(
      reinterpret_cast<const ushort*>(hp.data_ptr()),
      reinterpret_cast<const ushort*>(whT.data_ptr()),
      reinterpret_cast<const ushort*>(gx_t.data_ptr()),
      gx_t.stride(0),
      reinterpret_cast<const ushort*>(c_prev.data_ptr()),
      reinterpret_cast<ushort*>(s_out.data_ptr()),
      reinterpret_cast<ushort*>(c_out.data_ptr()),
      i_s.data_ptr<float>(), g_t.data_ptr<float>(), f_s.data_ptr<float>(),
      o_s.data_ptr<float>(), B, H, P, (float)forget_bias);
  HIP_CHECK_KERNEL();
}

// ---------------------------------------------------------------------------
// backward: one timestep, ds GEMM (dHP @ w_proj^T) + gate backward
// ---------------------------------------------------------------------------
// A = dHP [B, P] bf16; Bmat = w_proj [H, P] bf16 (its natural layout IS
// the transposed operand).  dG_t gets the 4 gate-grouped column slices;
// dc_prev the carry.  Strided upstream slice d_out is pre-combined into
// dHP by the caller.
template <int BB, int BH>
__global__ __launch_bounds__(BB * 4, 2)
void lstm_step_bwd_kernel(const ushort* __restrict__ dhp,
                          const ushort* __restrict__ wproj,
                          const ushort* __restrict__ dc_in,
                          const ushort* __restrict__ c_prev,
                          const ushort* __restrict__ c_out,
                          const float* __restrict__ i_s,
                          const float* __restrict__ g_t,
                          const float* __restrict__ f_s,
                          const float* __restrict__ o_s,
                          ushort* __restrict__ dgates,
                          ushort* __restrict__ dc_prev,
                          int Bsz, int H, int P) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  constexpr int THREADS = BB * 4;
  constexpr int A_BYTES = BB * LBK * 2;
  constexpr int B_BYTES = BH * LBK * 2;
  constexpr int PIECES_A = A_BYTES / (THREADS * 16);
  constexpr int PIECES_B = B_BYTES / (THREADS * 16);
  constexpr int NF = BH / 16;
#define LFA(b) (smem + (b) * (A_BYTES + B_BYTES))
#define LFB(b) (smem + (b) * (A_BYTES + B_BYTES) + A_BYTES)

  const int nbh = H / BH;
  const int j0 = (blockIdx.x % nbh) * BH;
  const int row0 = (blockIdx.x / nbh) * BB;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int fr = lane & 15;
  const int ko = (lane >> 4) * 8;

  f32x4 acc[NF] = {};

  int pa_row[PIECES_A], pa_kb[PIECES_A];
  long a_src_row[PIECES_A];
#pragma unroll
  for (int i = 0; i < PIECES_A; ++i) {
    const int p = (tid + i * THREADS) * 16;
    pa_row[i] = p / (LBK * 2);
    pa_kb[i] = (p % (LBK * 2)) ^ ((pa_row[i] & 7) << 4);
    long ar = row0 + pa_row[i];
    a_src_row[i] = ar < Bsz ? ar : Bsz - 1;
  }
  int pb_row[PIECES_B], pb_kb[PIECES_B];
#pragma unroll
  for (int i = 0; i < PIECES_B; ++i) {
    const int p = (tid + i * THREADS) * 16;
    pb_row[i] = p / (LBK * 2);
    pb_kb[i] = (p % (LBK * 2)) ^ ((pb_row[i] & 7) << 4);
  }
  const int ktiles = P / LBK;

#define LBSTAGE(buf, kt)                                                  \
  do {                                                                     \
    const long kbase = (long)(kt) * LBK;                                   \
    _Pragma("unroll") for (int i = 0; i < PIECES_A; ++i) {                 \
      const ushort* src = dhp + a_src_row[i] * P + kbase +                 \
                          (pa_kb[i] >> 1);                                 \
      __builtin_amdgcn_global_load_lds(                                    \
          (const __attribute__((address_space(1))) void*)src,              \
          (__attribute__((address_space(3))) void*)(LFA(buf) +             \
              (tid + i * THREADS) * 16), 16, 0, 0);                        \
    }                                                                      \
    _Pragma("unroll") for (int i = 0; i < PIECES_B; ++i) {                 \
      const ushort* src = wproj + (long)(j0 + pb_row[i]) * P + kbase +     \
                          (pb_kb[i] >> 1);                                 \
      __builtin_amdgcn_global_load_lds(                                    \
          (const __attribute__((address_space(1))) void*)src,              \
          (__attribute__((address_space(3))) void*)(LFB(buf) +             \
              (tid + i * THREADS) * 16), 16, 0, 0);                        \
    }                                                                      \
  } while (0)

  LBSTAGE(0, 0);
  __builtin_amdgcn_s_waitcnt(0);
  __builtin_amdgcn_s_barrier();
  int cur = 0;
  for (int kt = 0; kt < ktiles; ++kt) {
    if (kt + 1 < ktiles) LBSTAGE(cur ^ 1, kt + 1);
#pragma unroll
    for (int kk = 0; kk < LBK / 32; ++kk) {
      bf16x8 a_frag;
      {
        const int ar = wave * 16 + fr;
        a_frag = *(const bf16x8*)(LFA(cur) + lswz(ar, (kk * 32 + ko) * 2));
      }
#pragma unroll
      for (int nf = 0; nf < NF; ++nf) {
        const int br = nf * 16 + fr;
        bf16x8 b_frag =
            *(const bf16x8*)(LFB(cur) + lswz(br, (kk * 32 + ko) * 2));
        acc[nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag, b_frag, acc[nf], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_waitcnt(0);
    __builtin_amdgcn_s_barrier();
    cur ^= 1;
  }

  const int crow0 = (lane >> 4) * 4;
#pragma unroll
  for (int nf = 0; nf < NF; ++nf) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const long row = row0 + wave * 16 + crow0 + r;
      if (row >= Bsz) continue;
      const long col = j0 + nf * 16 + fr;
      const long hidx = row * H + col;
      const float dhv = acc[nf][r];        // ds value from the GEMM
      const float tc =
          tanhf(to_f32<__hip_bfloat16>(((const __hip_bfloat16*)c_out)[hidx]));
      const float ov = o_s[hidx];
      const float dov = dhv * tc;
      const float dc = dhv * ov * (1.f - tc * tc) +
          to_f32<__hip_bfloat16>(((const __hip_bfloat16*)dc_in)[hidx]);
      const float iv = i_s[hidx], gv = g_t[hidx], fv = f_s[hidx];
      const float div_ = dc * gv;
      const float dgv = dc * iv;
      const float dfv =
          dc * to_f32<__hip_bfloat16>(((const __hip_bfloat16*)c_prev)[hidx]);
      __hip_bfloat16 t0 = __float2bfloat16(dc * fv);
      dc_prev[hidx] = *reinterpret_cast<ushort*>(&t0);
      const long g0 = row * 4 * (long)H + col;
      __hip_bfloat16 t1 = __float2bfloat16(div_ * iv * (1.f - iv));
      dgates[g0] = *reinterpret_cast<ushort*>(&t1);
      __hip_bfloat16 t2 = __float2bfloat16(dgv * (1.f - gv * gv));
      dgates[g0 + H] = *reinterpret_cast<ushort*>(&t2);
      __hip_bfloat16 t3 = __float2bfloat16(dfv * fv * (1.f - fv));
      dgates[g0 + 2 * H] = *reinterpret_cast<ushort*>(&t3);
      __hip_bfloat16 t4 = __float2bfloat16(dov * ov * (1.f - ov));
      dgates[g0 + 3 * H] = *reinterpret_cast<ushort*>(&t4);
    }
  }
#undef LFA
#undef LFB
#undef LBSTAGE
}

void lstm_step_bwd_fused(const at::Tensor& dhp, const at::Tensor& wproj,
                         const at::Tensor& dc_in, const at::Tensor& c_prev,
                         const at::Tensor& c_out, const at::Tensor& i_s,
                         const at::Tensor& g_t, const at::Tensor& f_s,
                         const at::Tensor& o_s, at::Tensor dgates,
                         at::Tensor dc_prev) {
  TORCH_CHECK(dhp.is_cuda() && dhp.scalar_type() == at::kBFloat16 &&
              wproj.scalar_type() == at::kBFloat16);
  TORCH_CHECK(dhp.is_contiguous() && wproj.is_contiguous() &&
              dc_in.is_contiguous() && dgates.is_contiguous() &&
              dc_prev.is_contiguous());
  const int B = (int)dhp.size(0), P = (int)dhp.size(1);
  const int H = (int)wproj.size(0);
  TORCH_CHECK(wproj.size(1) == P && P % LBK == 0 && H % 32 == 0 &&
              B % 16 == 0 && B <= 128);
  constexpr int BB = 64, BH = 32;
  TORCH_CHECK(B % BB == 0, "lstm fused bwd: B must be a multiple of 64");
  const int blocks = (H / BH) * (B / BB);
  const int smem = 2 * (BB * LBK * 2 + BH * LBK * 2);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipFuncSetAttribute(
      reinterpret_cast<const void*>(&lstm_step_bwd_kernel<BB, BH>),
      hipFuncAttributeMaxDynamicSharedMemorySize, smem);
  lstm_step_bwd_kernel<BB, BH><<<blocks, BB * 4, smem, stream>>>(
      reinterpret_cast<const ushort*>(dhp.data_ptr()),
      reinterpret_cast<const ushort*>(wproj.data_ptr()),
      reinterpret_cast<const ushort*>(dc_in.data_ptr()),
      reinterpret_cast<const ushort*>(c_prev.data_ptr()),
      reinterpret_cast<const ushort*>(c_out.data_ptr()),
      i_s.data_ptr<float>(), g_t.data_ptr<float>(), f_s.data_ptr<float>(),
      o_s.data_ptr<float>(),
      reinterpret_cast<ushort*>(dgates.data_ptr()),
      reinterpret_cast<ushort*>(dc_prev.data_ptr()), B, H, P);
  HIP_CHECK_KERNEL();
}
