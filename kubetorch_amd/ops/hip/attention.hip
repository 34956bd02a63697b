// Fused causal attention FORWARD for MI355X (gfx950), bf16, head_dim=128.
//
// Replaces AOTriton's triton-compiled attn_fwd; backward stays on torch's
// AITER asm kernels (pairing in python — ops.flash_attention). Flash-style
// online softmax; MFMA via rocWMMA 16x16x32 fragments; the softmax runs
// fully in-register on the accumulator fragments using the gfx950 C/D
// mapping col = lane&15, row = (lane>>4)*4 + reg (verified empirically:
// LSE matches aten to 1e-6 — tests/test_ops.py::TestFlashAttention).
//
//   grid block = (batch, q_head, q-tile of 64 rows); 4 waves x 16 rows
//   per kv-tile (64 cols):
//     S = Q Kt            (4x 16x16 acc frags / wave, kept in registers)
//     online softmax      (in-register: per-lane rows r=0..3, 16-lane
//                          shuffle reduction across columns)
//     P -> LDS (bf16)     (only LDS hop: P must become a matrix_a frag)
//     O = O*rescale + P V (rescale in-register on the same row mapping)
//   outputs: O bf16, LSE = m + ln(l) fp32 [B,H,S] (aten layout).
//
// GQA native: kv head = q_head / (Hq/Hkv). S % 64 == 0 required.
#include <hip/hip_runtime.h>
#include <rocwmma/rocwmma.hpp>

using rocwmma::accumulator;
using rocwmma::col_major;
using rocwmma::matrix_a;
using rocwmma::matrix_b;
using rocwmma::row_major;

typedef unsigned short u16;
typedef ushort vec8u __attribute__((ext_vector_type(8)));

#define WAVE 64
#define QT 128     // q rows per workgroup (32 per wave = 2 row-blocks)
#define KT 64      // kv cols per tile
#define DH 128     // head dim
#define NWAVES 4
#define RB 2       // row-blocks of 16 per wave
#define DHP (DH + 8)  // padded LDS ld (breaks 32-way bank conflicts)
#define KTP (KT + 8)

using FragQ = rocwmma::fragment<matrix_a, 16, 16, 32, rocwmma::bfloat16_t, row_major>;
using FragKt = rocwmma::fragment<matrix_b, 16, 16, 32, rocwmma::bfloat16_t, col_major>;
using FragP = rocwmma::fragment<matrix_a, 16, 16, 32, rocwmma::bfloat16_t, row_major>;
using FragV = rocwmma::fragment<matrix_b, 16, 16, 32, rocwmma::bfloat16_t, row_major>;
using FragAcc = rocwmma::fragment<accumulator, 16, 16, 32, float>;

__device__ __forceinline__ u16 f2bf(float f) {
  unsigned int u = __float_as_uint(f);
  u += 0x7fff + ((u >> 16) & 1);
  return (u16)(u >> 16);
}

__global__ __launch_bounds__(256) void attn_fwd_kernel(
    const u16* __restrict__ q, const u16* __restrict__ k,
    const u16* __restrict__ v, u16* __restrict__ o, float* __restrict__ lse,
    int B, int Hq, int Hkv, int S, float scale) {
  const int n_qt = S / QT;
  int bid = blockIdx.x;
  const int qt = bid % n_qt;
  bid /= n_qt;
  const int hq = bid % Hq;
  const int b = bid / Hq;
  const int hkv = hq / (Hq / Hkv);
  const int q0 = qt * QT;

  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int acc_col = lane & 15;          // C/D fragment mapping
  const int acc_row0 = (lane >> 4) * 4;   // rows acc_row0..+3 in regs 0..3

  const size_t q_base = ((size_t)(b * Hq + hq) * S) * DH;
  const size_t kv_base = ((size_t)(b * Hkv + hkv) * S) * DH;

  __shared__ u16 k_lds[KT][DHP];
  __shared__ u16 v_lds[KT][DHP];
  __shared__ u16 p_lds[NWAVES][RB * 16][KTP];

  // each wave owns rows q0 + wid*32 + rb*16 + [0,16)
  FragQ fq[RB][4];
#pragma unroll
  for (int rb = 0; rb < RB; ++rb) {
    const u16* q_tile = q + q_base + (size_t)(q0 + wid * 32 + rb * 16) * DH;
#pragma unroll
    for (int c = 0; c < 4; ++c)
      rocwmma::load_matrix_sync(
          fq[rb][c], (const rocwmma::bfloat16_t*)(q_tile + c * 32), DH);
  }

  FragAcc facc[RB][8];
#pragma unroll
  for (int rb = 0; rb < RB; ++rb)
#pragma unroll
    for (int n = 0; n < 8; ++n) rocwmma::fill_fragment(facc[rb][n], 0.f);

  float m_run[RB][4], l_run[RB][4];
#pragma unroll
  for (int rb = 0; rb < RB; ++rb)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      m_run[rb][r] = -INFINITY;
      l_run[rb][r] = 0.f;
    }

  const int kv_end = q0 + QT;
  for (int kv0 = 0; kv0 < kv_end; kv0 += KT) {
    {  // cooperative K/V staging (rows padded to DHP)
      const u16* ksrc = k + kv_base + (size_t)kv0 * DH;
      const u16* vsrc = v + kv_base + (size_t)kv0 * DH;
      for (int t = threadIdx.x; t < KT * DH / 8; t += 256) {
        const int row = t / (DH / 8);
        const int col = (t % (DH / 8)) * 8;
        *(vec8u*)&k_lds[row][col] = ((const vec8u*)ksrc)[t];
        *(vec8u*)&v_lds[row][col] = ((const vec8u*)vsrc)[t];
      }
    }
    __syncthreads();

    // S = Q Kt : K fragments shared across both row-blocks
    FragAcc fs[RB][4];
#pragma unroll
    for (int n = 0; n < 4; ++n) {
#pragma unroll
      for (int rb = 0; rb < RB; ++rb) rocwmma::fill_fragment(fs[rb][n], 0.f);
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        FragKt fk;
        rocwmma::load_matrix_sync(
            fk, (const rocwmma::bfloat16_t*)(&k_lds[n * 16][c * 32]), DHP);
#pragma unroll
        for (int rb = 0; rb < RB; ++rb)
          rocwmma::mma_sync(fs[rb][n], fq[rb][c], fk, fs[rb][n]);
      }
    }

    // in-register online softmax (per-lane rows acc_row0+r per row-block)
    float resc[RB][4];
#pragma unroll
    for (int rb = 0; rb < RB; ++rb) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qi = q0 + wid * 32 + rb * 16 + acc_row0 + r;
        float mloc = -INFINITY;
        float sv[4];
#pragma unroll
        for (int n = 0; n < 4; ++n) {
          const int kj = kv0 + n * 16 + acc_col;
          float val = (kj <= qi) ? fs[rb][n][r] * scale : -INFINITY;
          sv[n] = val;
          mloc = fmaxf(mloc, val);
        }
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
          mloc = fmaxf(mloc, __shfl_xor(mloc, off, WAVE));
        const float m_new = fmaxf(m_run[rb][r], mloc);
        float psum = 0.f;
#pragma unroll
        for (int n = 0; n < 4; ++n) {
          float pp = (sv[n] == -INFINITY) ? 0.f : __expf(sv[n] - m_new);
          p_lds[wid][rb * 16 + acc_row0 + r][n * 16 + acc_col] = f2bf(pp);
          psum += pp;
        }
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
          psum += __shfl_xor(psum, off, WAVE);
        resc[rb][r] =
            (m_run[rb][r] == -INFINITY) ? 0.f : __expf(m_run[rb][r] - m_new);
        l_run[rb][r] = l_run[rb][r] * resc[rb][r] + psum;
        m_run[rb][r] = m_new;
      }
    }

    // O = O*rescale + P V : V fragments shared across both row-blocks
#pragma unroll
    for (int rb = 0; rb < RB; ++rb)
#pragma unroll
      for (int n = 0; n < 8; ++n)
#pragma unroll
        for (int r = 0; r < 4; ++r) facc[rb][n][r] *= resc[rb][r];
    FragP fp[RB][2];
#pragma unroll
    for (int rb = 0; rb < RB; ++rb) {
      rocwmma::load_matrix_sync(
          fp[rb][0], (const rocwmma::bfloat16_t*)&p_lds[wid][rb * 16][0], KTP);
      rocwmma::load_matrix_sync(
          fp[rb][1], (const rocwmma::bfloat16_t*)&p_lds[wid][rb * 16][32], KTP);
    }
#pragma unroll
    for (int n = 0; n < 8; ++n) {
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        FragV fv;
        rocwmma::load_matrix_sync(
            fv, (const rocwmma::bfloat16_t*)(&v_lds[c * 32][n * 16]), DHP);
#pragma unroll
        for (int rb = 0; rb < RB; ++rb)
          rocwmma::mma_sync(facc[rb][n], fp[rb][c], fv, facc[rb][n]);
      }
    }
    __syncthreads();  // K/V tile reuse barrier
  }

  // epilogue: O /= l in-register, stage each 16x16 block through LDS
  // (reusing p_lds as float scratch) to convert fp32 -> bf16 coalesced.
#pragma unroll
  for (int rb = 0; rb < RB; ++rb) {
    float inv[4];
#pragma unroll
    for (int r = 0; r < 4; ++r)
      inv[r] = (l_run[rb][r] > 0.f) ? 1.f / l_run[rb][r] : 0.f;
    u16* o_tile = o + q_base + (size_t)(q0 + wid * 32 + rb * 16) * DH;
    float* scratch = (float*)&p_lds[wid][0][0];  // 16x16 f32 scratch
#pragma unroll
    for (int n = 0; n < 8; ++n) {
#pragma unroll
      for (int r = 0; r < 4; ++r) facc[rb][n][r] *= inv[r];
      rocwmma::store_matrix_sync(scratch, facc[rb][n], 16,
                                 rocwmma::mem_row_major);
      __syncthreads();
      const int rr = lane >> 2;
      const int cc = (lane & 3) * 4;
#pragma unroll
      for (int j = 0; j < 4; ++j)
        o_tile[(size_t)rr * DH + n * 16 + cc + j] =
            f2bf(scratch[rr * 16 + cc + j]);
      __syncthreads();
    }
    if (acc_col == 0) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qi = q0 + wid * 32 + rb * 16 + acc_row0 + r;
        lse[((size_t)(b * Hq + hq)) * S + qi] =
            m_run[rb][r] + __logf(fmaxf(l_run[rb][r], 1e-30f));
      }
    }
  }
}

extern "C" void kt_attn_fwd(const void* q, const void* k, const void* v,
                            void* o, void* lse, int B, int Hq, int Hkv,
                            int S, float scale, hipStream_t stream) {
  const int n_qt = S / QT;
  dim3 grid(B * Hq * n_qt);
  hipLaunchKernelGGL(attn_fwd_kernel, grid, dim3(256), 0, stream,
                     (const u16*)q, (const u16*)k, (const u16*)v, (u16*)o,
                     (float*)lse, B, Hq, Hkv, S, scale);
}
