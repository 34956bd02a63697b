// Fused causal attention FORWARD for MI355X (gfx950), bf16, head_dim=128.
//
// Replaces AOTriton's triton-compiled attn_fwd (~330 TF on this shape);
// backward stays on torch's AITER asm kernels (the autograd pairing happens
// in python — see ops.flash_attention). Structure per the CDNA guide's
// attention recipe (flash-style online softmax, LDS-staged K/V, MFMA via
// rocWMMA 16x16x32 fragments, 4 waves x 16 q-rows per workgroup):
//
//   grid block = (batch, q_head, q_tile of 64 rows)
//   per kv-tile (64 cols):
//     S = scale * Q Kt   (4x 16x16 frags / wave -> LDS scratch)
//     online softmax in LDS (64 lanes = 16 rows x 4 col-groups)
//     P (bf16, LDS) x V -> O fragments, rescaled in-register using the
//     verified gfx950 C/D mapping col=lane&15, row=(lane>>4)*4+reg.
//   outputs: O bf16 and LSE = m + ln(l) (fp32, aten layout [B,H,S]).
//
// GQA native: kv head = q_head / (Hq/Hkv).
#include <hip/hip_runtime.h>
// note: rocwmma::bfloat16_t (hip_bfloat16) is bit-compatible with torch bf16
#include <rocwmma/rocwmma.hpp>

using rocwmma::accumulator;
using rocwmma::col_major;
using rocwmma::matrix_a;
using rocwmma::matrix_b;
using rocwmma::row_major;

typedef unsigned short u16;
typedef ushort vec8u __attribute__((ext_vector_type(8)));

#define WAVE 64
#define QT 64     // q rows per workgroup
#define KT 64     // kv cols per tile
#define DH 128    // head dim
#define NWAVES 4  // QT/16

using FragQ = rocwmma::fragment<matrix_a, 16, 16, 32, rocwmma::bfloat16_t, row_major>;
using FragKt = rocwmma::fragment<matrix_b, 16, 16, 32, rocwmma::bfloat16_t, col_major>;
using FragP = rocwmma::fragment<matrix_a, 16, 16, 32, rocwmma::bfloat16_t, row_major>;
using FragV = rocwmma::fragment<matrix_b, 16, 16, 32, rocwmma::bfloat16_t, row_major>;
using FragAcc = rocwmma::fragment<accumulator, 16, 16, 32, float>;

__global__ __launch_bounds__(256) void attn_fwd_kernel(
    const u16* __restrict__ q, const u16* __restrict__ k,
    const u16* __restrict__ v, u16* __restrict__ o, float* __restrict__ lse,
    int B, int Hq, int Hkv, int S, float scale) {
  // block -> (b, hq, q-tile)
  const int n_qt = S / QT;
  int bid = blockIdx.x;
  const int qt = bid % n_qt;
  bid /= n_qt;
  const int hq = bid % Hq;
  const int b = bid / Hq;
  const int hkv = hq / (Hq / Hkv);
  const int q0 = qt * QT;

  const int wid = threadIdx.x >> 6;   // wave 0..3; owns q rows q0+wid*16..+16
  const int lane = threadIdx.x & 63;

  const size_t q_base = ((size_t)(b * Hq + hq) * S) * DH;
  const size_t kv_base = ((size_t)(b * Hkv + hkv) * S) * DH;

  // LDS: K tile, V tile, per-wave S scratch + P tile, softmax state
  __shared__ u16 k_lds[KT][DH];            // 16 KB
  __shared__ u16 v_lds[KT][DH];            // 16 KB
  __shared__ float s_lds[NWAVES][16][KT];  // 16 KB
  __shared__ u16 p_lds[NWAVES][16][KT];    // 8 KB
  __shared__ float o_scale[NWAVES][16];    // per-row O rescale factor

  // Q fragments: 4 k-chunks of 32, loop-invariant
  FragQ fq[4];
  const u16* q_tile = q + q_base + (size_t)(q0 + wid * 16) * DH;
#pragma unroll
  for (int c = 0; c < 4; ++c)
    rocwmma::load_matrix_sync(fq[c],
                              (const rocwmma::bfloat16_t*)(q_tile + c * 32), DH);

  FragAcc facc[8];  // O accumulator: 8 col-blocks of 16
#pragma unroll
  for (int n = 0; n < 8; ++n) rocwmma::fill_fragment(facc[n], 0.f);

  // online softmax state for the wave's 16 rows (each row owned by 4 lanes:
  // lane = row*4 + grp, grp covers cols grp*16..+16 of the kv tile)
  const int srow = lane >> 2;   // 0..15
  const int sgrp = lane & 3;    // 0..3
  float m_run = -INFINITY;      // valid in all 4 lanes of the row group
  float l_run = 0.f;

  const int kv_end = q0 + QT;   // causal: kv tiles up to the q-tile end
  for (int kv0 = 0; kv0 < kv_end; kv0 += KT) {
    // cooperative K/V tile staging (coalesced: 256 threads x 16B)
    {
      const u16* ksrc = k + kv_base + (size_t)kv0 * DH;
      const u16* vsrc = v + kv_base + (size_t)kv0 * DH;
      // KT*DH = 8192 elems = 1024 vec8; 256 threads -> 4 vecs each
      for (int t = threadIdx.x; t < KT * DH / 8; t += 256) {
        ((vec8u*)k_lds)[t] = ((const vec8u*)ksrc)[t];
        ((vec8u*)v_lds)[t] = ((const vec8u*)vsrc)[t];
      }
    }
    __syncthreads();

    // S = Q Kt : 4 col-frags x 4 k-chunks
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      FragAcc fs;
      rocwmma::fill_fragment(fs, 0.f);
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        FragKt fk;
        rocwmma::load_matrix_sync(
            fk, (const rocwmma::bfloat16_t*)(&k_lds[n * 16][c * 32]), DH);
        rocwmma::mma_sync(fs, fq[c], fk, fs);
      }
      rocwmma::store_matrix_sync(&s_lds[wid][0][n * 16], fs, KT,
                                 rocwmma::mem_row_major);
    }
    __syncthreads();

    // online softmax on the wave's 16x64 score block
    {
      const int qi = q0 + wid * 16 + srow;  // global q row
      float mloc = -INFINITY;
      float sv[16];
#pragma unroll
      for (int j = 0; j < 16; ++j) {
        const int kj = kv0 + sgrp * 16 + j;
        float val = (kj <= qi) ? s_lds[wid][srow][sgrp * 16 + j] * scale
                               : -INFINITY;
        sv[j] = val;
        mloc = fmaxf(mloc, val);
      }
      // row max across the 4 col-groups (lanes row*4..row*4+3)
#pragma unroll
      for (int off = 1; off < 4; off <<= 1)
        mloc = fmaxf(mloc, __shfl_xor(mloc, off, WAVE));
      const float m_new = fmaxf(m_run, mloc);
      float psum = 0.f;
#pragma unroll
      for (int j = 0; j < 16; ++j) {
        float p = (sv[j] == -INFINITY) ? 0.f : __expf(sv[j] - m_new);
        unsigned int u = __float_as_uint(p);
        u += 0x7fff + ((u >> 16) & 1);  // rne bf16
        p_lds[wid][srow][sgrp * 16 + j] = (u16)(u >> 16);
        psum += p;
      }
#pragma unroll
      for (int off = 1; off < 4; off <<= 1)
        psum += __shfl_xor(psum, off, WAVE);
      const float rescale = (m_run == -INFINITY) ? 0.f : __expf(m_run - m_new);
      l_run = l_run * rescale + psum;
      m_run = m_new;
      if (sgrp == 0) o_scale[wid][srow] = rescale;
    }
    __syncthreads();

    // O = O*rescale + P V  (rescale in-register: gfx950 C/D mapping
    // col = lane&15, row = (lane>>4)*4 + reg)
    {
      const int acc_col = lane & 15;
      const int acc_row0 = (lane >> 4) * 4;
#pragma unroll
      for (int n = 0; n < 8; ++n) {
#pragma unroll
        for (int r = 0; r < 4; ++r)
          facc[n].x[r] *= o_scale[wid][acc_row0 + r];
      }
      (void)acc_col;
      FragP fp[2];
      rocwmma::load_matrix_sync(fp[0], (const rocwmma::bfloat16_t*)&p_lds[wid][0][0], KT);
      rocwmma::load_matrix_sync(fp[1], (const rocwmma::bfloat16_t*)&p_lds[wid][0][32], KT);
#pragma unroll
      for (int n = 0; n < 8; ++n) {
#pragma unroll
        for (int c = 0; c < 2; ++c) {
          FragV fv;
          rocwmma::load_matrix_sync(
              fv, (const rocwmma::bfloat16_t*)(&v_lds[c * 32][n * 16]), DH);
          rocwmma::mma_sync(facc[n], fp[c], fv, facc[n]);
        }
      }
    }
    __syncthreads();  // K/V tile reuse barrier
  }

  // epilogue: O / l, write bf16 O and fp32 LSE
  {
    // stage O through s_lds scratch per 16-col block (mapping-free store)
    u16* o_tile = o + q_base + (size_t)(q0 + wid * 16) * DH;
    const float l_inv = (l_run > 0.f) ? 1.f / l_run : 0.f;
    // broadcast per-row 1/l into LDS
    if (sgrp == 0) o_scale[wid][srow] = l_inv;
    __syncthreads();
    float* scratch = &s_lds[wid][0][0];  // flat 16x16 (ld=16) scratch
#pragma unroll
    for (int n = 0; n < 8; ++n) {
      rocwmma::store_matrix_sync(scratch, facc[n], 16, rocwmma::mem_row_major);
      __syncthreads();
      // 64 lanes write the 16x16 block: lane -> (row=lane>>2, 4 cols)
      const int rr = lane >> 2;
      const int cc = (lane & 3) * 4;
      float inv = o_scale[wid][rr];
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float val = scratch[rr * 16 + cc + j] * inv;
        unsigned int u = __float_as_uint(val);
        u += 0x7fff + ((u >> 16) & 1);
        o_tile[(size_t)rr * DH + n * 16 + cc + j] = (u16)(u >> 16);
      }
      __syncthreads();
    }
    if (sgrp == 0) {
      const int qi = q0 + wid * 16 + srow;
      lse[((size_t)(b * Hq + hq)) * S + qi] =
          m_run + __logf(fmaxf(l_run, 1e-30f));
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
