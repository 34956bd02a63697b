// MI355X (gfx950 / CDNA4) kernels for the kubetorch_amd compute path.
//
// Design notes (see /opt/skills/guides/cdna_hip_programming.md):
//  - wave = 64 lanes; block sizes are multiples of 64 (256 default).
//  - all bf16 global traffic is vectorized as ushort8 (16 B/lane) --
//    hipcc does NOT auto-vectorize scalar bf16 loads (Guideline 13).
//  - memory-bound kernels cap the grid at ~2048 blocks and grid-stride.
//  - reductions: wave shuffle (width 64) -> LDS across waves.
//  - RoPE uses a host-precomputed cos/sin table (no device trig).
//
// These kernels replace the multi-kernel eager-PyTorch sequences for the
// hot memory-bound ops of the Llama training step (RMSNorm, RoPE, SwiGLU,
// fused cross-entropy, fused AdamW). GEMMs stay on hipBLASLt via torch.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

typedef unsigned short u16;
typedef ushort vec8u __attribute__((ext_vector_type(8)));
typedef ushort vec4u __attribute__((ext_vector_type(4)));
typedef float vec4f __attribute__((ext_vector_type(4)));

#define WAVE 64

__device__ __forceinline__ float bf2f(u16 x) {
  unsigned int u = ((unsigned int)x) << 16;
  return __uint_as_float(u);
}
__device__ __forceinline__ u16 f2bf(float f) {
  // round-to-nearest-even bf16 conversion
  unsigned int u = __float_as_uint(f);
  unsigned int rounding = 0x7fff + ((u >> 16) & 1);
  u += rounding;
  return (u16)(u >> 16);
}

// Block-wide sum; every thread returns the total. Reusable across
// iterations (trailing barrier protects the LDS scratch).
__device__ __forceinline__ float block_reduce_sum(float v) {
  __shared__ float s[16];
#pragma unroll
  for (int o = 32; o > 0; o >>= 1) v += __shfl_down(v, o, WAVE);
  const int wid = threadIdx.x >> 6;
  const int nw = blockDim.x >> 6;
  if ((threadIdx.x & 63) == 0) s[wid] = v;
  __syncthreads();
  float t = 0.f;
  for (int i = 0; i < nw; ++i) t += s[i];
  __syncthreads();
  return t;
}

// ---------------------------------------------------------------------------
// (Fused-residual) RMSNorm forward.
//   s = x (+ res, when res != null); y = s * rsqrt(mean(s^2)+eps) * w
// When res is given, the summed residual stream s is written to s_out —
// fusing the transformer's pre-norm residual add into the norm kernel
// (saves a full elementwise pass per norm). One block per row; s re-read
// in pass 2 (hot in L2; register caching would spill — see guide rule 20).
// ---------------------------------------------------------------------------
__global__ void rmsnorm_fwd_kernel(const u16* __restrict__ x,
                                   const u16* __restrict__ res,
                                   const u16* __restrict__ w,
                                   u16* __restrict__ y,
                                   u16* __restrict__ s_out,
                                   float* __restrict__ invrms,
                                   int N, int H, float eps) {
  const int nvec = H >> 3;
  for (int row = blockIdx.x; row < N; row += gridDim.x) {
    const u16* xr = x + (size_t)row * H;
    const u16* rr = res ? res + (size_t)row * H : nullptr;
    u16* yr = y + (size_t)row * H;
    u16* sr = s_out ? s_out + (size_t)row * H : nullptr;
    float ss = 0.f;
    for (int vI = threadIdx.x; vI < nvec; vI += blockDim.x) {
      vec8u xv = *reinterpret_cast<const vec8u*>(xr + vI * 8);
      vec8u sv;
      if (rr) {
        vec8u rv = *reinterpret_cast<const vec8u*>(rr + vI * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) sv[j] = f2bf(bf2f(xv[j]) + bf2f(rv[j]));
        if (sr) *reinterpret_cast<vec8u*>(sr + vI * 8) = sv;
      } else {
        sv = xv;
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf2f(sv[j]);
        ss += f * f;
      }
    }
    ss = block_reduce_sum(ss);
    const float ir = rsqrtf(ss / (float)H + eps);
    if (threadIdx.x == 0 && invrms) invrms[row] = ir;
    const u16* src = (rr && sr) ? sr : xr;  // pass 2 reads the summed stream
    for (int vI = threadIdx.x; vI < nvec; vI += blockDim.x) {
      vec8u xv = *reinterpret_cast<const vec8u*>(src + vI * 8);
      vec8u wv = *reinterpret_cast<const vec8u*>(w + vI * 8);
      vec8u ov;
#pragma unroll
      for (int j = 0; j < 8; ++j) ov[j] = f2bf(bf2f(xv[j]) * ir * bf2f(wv[j]));
      *reinterpret_cast<vec8u*>(yr + vI * 8) = ov;
    }
    __syncthreads();  // sr written this row must not race the next row
  }
}

// ---------------------------------------------------------------------------
// RMSNorm backward.
//   dx_i = ir*(dy_i*w_i) - x_i * ir^3/H * S,  S = sum_j dy_j*w_j*x_j
//   dw_j = sum_rows dy_j * x_j * ir   (accumulated per-block in LDS, written
//          to a [gridDim.x, H] fp32 partial buffer; reduced by colsum below)
// LDS budget: H fp32 <= 160KB -> H <= 40960 (Llama H=4096 -> 16 KB). Fine.
// ---------------------------------------------------------------------------
// ds (nullable): upstream gradient of the summed residual stream s (fused
// path) — added into dx, so dx = ds + d(norm)/ds·dy, which is the gradient
// for BOTH inputs of the fused add (they are identical).
__global__ void rmsnorm_bwd_kernel(const u16* __restrict__ dy,
                                   const u16* __restrict__ ds,
                                   const u16* __restrict__ x,
                                   const u16* __restrict__ w,
                                   const float* __restrict__ invrms,
                                   u16* __restrict__ dx,
                                   float* __restrict__ dw_partial,
                                   int N, int H) {
  // dw accumulator in LDS, stored COLUMN-major (dwacc[j*nvec + vI]) so a
  // wave's slot-j writes hit consecutive words -> conflict-free (the
  // row-major [vI*8+j] layout was a 16-way conflict: 5.9e9
  // SQ_LDS_BANK_CONFLICT per bench step, profiles/r01_pmc.md).
  extern __shared__ float dwacc[];  // [8][nvec]
  const int nvec = H >> 3;
  for (int t = threadIdx.x; t < H; t += blockDim.x) dwacc[t] = 0.f;
  __syncthreads();
  for (int row = blockIdx.x; row < N; row += gridDim.x) {
    const u16* dyr = dy + (size_t)row * H;
    const u16* xr = x + (size_t)row * H;
    u16* dxr = dx + (size_t)row * H;
    const float ir = invrms[row];
    float S = 0.f;
    for (int vI = threadIdx.x; vI < nvec; vI += blockDim.x) {
      vec8u dyv = *reinterpret_cast<const vec8u*>(dyr + vI * 8);
      vec8u wv = *reinterpret_cast<const vec8u*>(w + vI * 8);
      vec8u xv = *reinterpret_cast<const vec8u*>(xr + vI * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) S += bf2f(dyv[j]) * bf2f(wv[j]) * bf2f(xv[j]);
    }
    S = block_reduce_sum(S);
    const float k = ir * ir * ir * S / (float)H;
    const u16* dsr = ds ? ds + (size_t)row * H : nullptr;
    for (int vI = threadIdx.x; vI < nvec; vI += blockDim.x) {
      vec8u dyv = *reinterpret_cast<const vec8u*>(dyr + vI * 8);
      vec8u wv = *reinterpret_cast<const vec8u*>(w + vI * 8);
      vec8u xv = *reinterpret_cast<const vec8u*>(xr + vI * 8);
      vec8u dsv;
      if (dsr) dsv = *reinterpret_cast<const vec8u*>(dsr + vI * 8);
      vec8u ov;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float dyf = bf2f(dyv[j]);
        float xf = bf2f(xv[j]);
        float g = ir * dyf * bf2f(wv[j]) - xf * k;
        if (dsr) g += bf2f(dsv[j]);
        ov[j] = f2bf(g);
        dwacc[j * nvec + vI] += dyf * xf * ir;  // column-major: own slot,
                                                // consecutive lanes/banks
      }
      *reinterpret_cast<vec8u*>(dxr + vI * 8) = ov;
    }
    __syncthreads();
  }
  float* out = dw_partial + (size_t)blockIdx.x * H;
  for (int vI = threadIdx.x; vI < nvec; vI += blockDim.x) {
#pragma unroll
    for (int j = 0; j < 8; ++j) out[vI * 8 + j] = dwacc[j * nvec + vI];
  }
}

// Column-sum of the [P, H] fp32 partial buffer -> dw [H] bf16.
// One block per 4-column group (H/4 blocks keep the chip busy; the previous
// thread-per-column layout launched only H/1024 blocks and left 255/256 CUs
// idle), threads split the P rows, block-reduce per column.
__global__ void colsum_bf16_kernel(const float* __restrict__ partial,
                                   u16* __restrict__ out, int P, int H) {
  const int col = blockIdx.x * 4;
  if (col >= H) return;
  float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
  for (int p = threadIdx.x; p < P; p += blockDim.x) {
    vec4f v = *reinterpret_cast<const vec4f*>(partial + (size_t)p * H + col);
    a0 += v[0]; a1 += v[1]; a2 += v[2]; a3 += v[3];
  }
  a0 = block_reduce_sum(a0);
  a1 = block_reduce_sum(a1);
  a2 = block_reduce_sum(a2);
  a3 = block_reduce_sum(a3);
  if (threadIdx.x == 0) {
    vec4u o;
    o[0] = f2bf(a0); o[1] = f2bf(a1); o[2] = f2bf(a2); o[3] = f2bf(a3);
    *reinterpret_cast<vec4u*>(out + col) = o;
  }
}

// ---------------------------------------------------------------------------
// RoPE (Llama rotate-half), in-place capable, fwd/bwd via sign.
// x: [T, Hh, D] bf16 where T = B*S tokens (row-major), cos/sin: [S, D/2] fp32
// host-precomputed (Appendix B: no on-device trig). Each thread rotates 4
// pairs: 8 B from each half of the head dim.
//   fwd: o1 = x1*c - x2*s ; o2 = x2*c + x1*s      (sign=+1)
//   bwd: o1 = x1*c + x2*s ; o2 = x2*c - x1*s      (sign=-1)
// ---------------------------------------------------------------------------
__global__ void rope_kernel(const u16* __restrict__ x, u16* __restrict__ o,
                            const float* __restrict__ cost,
                            const float* __restrict__ sint,
                            long total_quads, int S, int Hh, int D,
                            long src_t_stride, long src_h_stride,
                            float sign, float oscale) {
  // oscale: free fp32 output scaling (applied before the bf16 round), used
  // to fold the attention softmax scale * log2e into Q for the CK v3 FMHA
  // kernel's scaled-log2-domain LSE contract (see bindings.cpp attn_fwd_v3).
  // x is read with explicit (token, head) strides so qkv-split views and
  // transposed gradients need no .contiguous() copy; o is written packed.
  const int half = D >> 1;
  const int quads_per_head = D >> 3;  // 4 pairs per quad
  const long quads_per_tok = (long)Hh * quads_per_head;
  for (long q = (long)blockIdx.x * blockDim.x + threadIdx.x; q < total_quads;
       q += (long)gridDim.x * blockDim.x) {
    const long tok = q / quads_per_tok;
    const int rem = (int)(q - tok * quads_per_tok);
    const int h = rem / quads_per_head;
    const int qi = rem - h * quads_per_head;
    const int pos = (int)(tok % S);
    const size_t base = ((size_t)tok * Hh + h) * D + qi * 4;
    const size_t sbase = (size_t)tok * src_t_stride +
                         (size_t)h * src_h_stride + qi * 4;
    vec4u x1 = *reinterpret_cast<const vec4u*>(x + sbase);
    vec4u x2 = *reinterpret_cast<const vec4u*>(x + sbase + half);
    vec4f c = *reinterpret_cast<const vec4f*>(cost + (size_t)pos * half + qi * 4);
    vec4f s = *reinterpret_cast<const vec4f*>(sint + (size_t)pos * half + qi * 4);
    vec4u o1, o2;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float a = bf2f(x1[j]), b = bf2f(x2[j]);
      o1[j] = f2bf((a * c[j] - sign * b * s[j]) * oscale);
      o2[j] = f2bf((b * c[j] + sign * a * s[j]) * oscale);
    }
    // NOT nontemporal: rope's K/V outputs are ~32 MB/layer and re-read
    // immediately by the attention kernel — keeping them L2-retained
    // measured faster (20,773 vs 20,442 tok/s with streaming stores)
    *reinterpret_cast<vec4u*>(o + base) = o1;
    *reinterpret_cast<vec4u*>(o + base + half) = o2;
  }
}

// ---------------------------------------------------------------------------
// SwiGLU: out = silu(gate) * up, with gate_up packed [N, 2I] from one GEMM.
// ---------------------------------------------------------------------------
__global__ void swiglu_fwd_kernel(const u16* __restrict__ gu,
                                  u16* __restrict__ out, long N, int I) {
  const long total = N * (I >> 3);
  const int nvec = I >> 3;
  for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += (long)gridDim.x * blockDim.x) {
    const long row = t / nvec;
    const int vI = (int)(t - row * nvec);
    const size_t gbase = (size_t)row * 2 * I + vI * 8;
    vec8u g = __builtin_nontemporal_load(
        reinterpret_cast<const vec8u*>(gu + gbase));
    vec8u u = __builtin_nontemporal_load(
        reinterpret_cast<const vec8u*>(gu + gbase + I));
    vec8u o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf2f(g[j]);
      float sig = 1.f / (1.f + __expf(-gf));
      o[j] = f2bf(gf * sig * bf2f(u[j]));
    }
    __builtin_nontemporal_store(
        o, reinterpret_cast<vec8u*>(out + (size_t)row * I + vI * 8));
  }
}

__global__ void swiglu_bwd_kernel(const u16* __restrict__ dout,
                                  const u16* __restrict__ gu,
                                  u16* __restrict__ dgu, long N, int I) {
  const long total = N * (I >> 3);
  const int nvec = I >> 3;
  for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += (long)gridDim.x * blockDim.x) {
    const long row = t / nvec;
    const int vI = (int)(t - row * nvec);
    const size_t gbase = (size_t)row * 2 * I + vI * 8;
    vec8u g = __builtin_nontemporal_load(
        reinterpret_cast<const vec8u*>(gu + gbase));
    vec8u u = __builtin_nontemporal_load(
        reinterpret_cast<const vec8u*>(gu + gbase + I));
    vec8u dov = __builtin_nontemporal_load(
        reinterpret_cast<const vec8u*>(dout + (size_t)row * I + vI * 8));
    vec8u dg, du;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf2f(g[j]);
      float uf = bf2f(u[j]);
      float dof = bf2f(dov[j]);
      float sig = 1.f / (1.f + __expf(-gf));
      float silu = gf * sig;
      dg[j] = f2bf(dof * uf * sig * (1.f + gf * (1.f - sig)));
      du[j] = f2bf(dof * silu);
    }
    __builtin_nontemporal_store(dg, reinterpret_cast<vec8u*>(dgu + gbase));
    __builtin_nontemporal_store(du,
                                reinterpret_cast<vec8u*>(dgu + gbase + I));
  }
}

// ---------------------------------------------------------------------------
// Fused cross entropy over bf16 logits [N, V]: computes per-row loss and
// OVERWRITES logits with d(mean loss)/dlogits = (softmax - onehot) * scale.
// Avoids materializing an fp32 softmax of the [N, 128256] logits (the torch
// eager path costs ~5x the HBM traffic). One block per row; pass 1 = online
// max+sumexp; pass 2 = grad write. ignore_index rows get loss 0 / grad 0.
// ---------------------------------------------------------------------------
__global__ void cross_entropy_fwd_kernel(u16* __restrict__ logits,
                                         const long* __restrict__ targets,
                                         float* __restrict__ loss,
                                         int N, int V, float scale,
                                         long ignore_index) {
  __shared__ float s_m[16], s_s[16];
  const int nvec = V >> 3;
  for (int row = blockIdx.x; row < N; row += gridDim.x) {
    u16* lr = logits + (size_t)row * V;
    const long tgt = targets[row];
    if (tgt == ignore_index) {
      for (int vI = threadIdx.x; vI < nvec; vI += blockDim.x) {
        vec8u z = {0, 0, 0, 0, 0, 0, 0, 0};
        *reinterpret_cast<vec8u*>(lr + vI * 8) = z;
      }
      if (threadIdx.x == 0) loss[row] = 0.f;
      __syncthreads();
      continue;
    }
    // pass 1: thread-local online max + sumexp
    float m = -INFINITY, s = 0.f;
    for (int vI = threadIdx.x; vI < nvec; vI += blockDim.x) {
      vec8u xv = *reinterpret_cast<const vec8u*>(lr + vI * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf2f(xv[j]);
        if (f > m) {
          s = s * __expf(m - f) + 1.f;
          m = f;
        } else {
          s += __expf(f - m);
        }
      }
    }
    // wave merge of (m, s); threads with no elements carry (m=-inf, s=0)
    // and merging two of those must not compute exp(-inf - -inf) = NaN.
#pragma unroll
    for (int o = 32; o > 0; o >>= 1) {
      float m2 = __shfl_down(m, o, WAVE);
      float s2 = __shfl_down(s, o, WAVE);
      float mn = fmaxf(m, m2);
      if (mn > -INFINITY) s = s * __expf(m - mn) + s2 * __expf(m2 - mn);
      m = mn;
    }
    const int wid = threadIdx.x >> 6;
    const int nw = blockDim.x >> 6;
    if ((threadIdx.x & 63) == 0) { s_m[wid] = m; s_s[wid] = s; }
    __syncthreads();
    float M = -INFINITY;
    for (int i = 0; i < nw; ++i) M = fmaxf(M, s_m[i]);
    float Z = 0.f;
    for (int i = 0; i < nw; ++i)
      if (s_m[i] > -INFINITY) Z += s_s[i] * __expf(s_m[i] - M);
    __syncthreads();
    const float logZ = __logf(Z) + M;
    if (threadIdx.x == 0) loss[row] = logZ - bf2f(lr[tgt]);
    __syncthreads();  // target logit must be read before pass 2 overwrites it
    const float inv_Z = 1.f / Z;
    // pass 2: grad in place
    for (int vI = threadIdx.x; vI < nvec; vI += blockDim.x) {
      vec8u xv = *reinterpret_cast<const vec8u*>(lr + vI * 8);
      vec8u ov;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const long idx = (long)vI * 8 + j;
        float p = __expf(bf2f(xv[j]) - M) * inv_Z;
        if (idx == tgt) p -= 1.f;
        ov[j] = f2bf(p * scale);
      }
      *reinterpret_cast<vec8u*>(lr + vI * 8) = ov;
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// Fused AdamW over a flat bf16 param/grad bucket with fp32 m, v state.
// One kernel per bucket (buckets are the DDP comm buckets), fp32 math,
// decoupled weight decay. grad_scale folds in any 1/world_size factor.
// Traffic: 22 B/param (vs ~5 separate eager kernels).
// ---------------------------------------------------------------------------
__global__ void adamw_kernel(u16* __restrict__ p, const u16* __restrict__ g,
                             float* __restrict__ m, float* __restrict__ v,
                             long n, float lr, float beta1, float beta2,
                             float eps, float wd, float bc1, float bc2,
                             float grad_scale) {
  // Pure streaming op (every byte touched exactly once): nontemporal
  // loads/stores bypass L2 retention so the 22 B/param of HBM traffic
  // doesn't thrash the cache the overlapped backward is using, and the
  // stores avoid read-for-ownership (measured 4.6 TB/s before).
  const long nvec = n >> 3;
  for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < nvec;
       t += (long)gridDim.x * blockDim.x) {
    const size_t i = (size_t)t * 8;
    vec8u pv = __builtin_nontemporal_load(reinterpret_cast<const vec8u*>(p + i));
    vec8u gv = __builtin_nontemporal_load(reinterpret_cast<const vec8u*>(g + i));
    vec4f m0 = __builtin_nontemporal_load(reinterpret_cast<const vec4f*>(m + i));
    vec4f m1 = __builtin_nontemporal_load(reinterpret_cast<const vec4f*>(m + i + 4));
    vec4f v0 = __builtin_nontemporal_load(reinterpret_cast<const vec4f*>(v + i));
    vec4f v1 = __builtin_nontemporal_load(reinterpret_cast<const vec4f*>(v + i + 4));
    vec8u po;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf2f(gv[j]) * grad_scale;
      float pf = bf2f(pv[j]);
      float mj = (j < 4) ? m0[j] : m1[j - 4];
      float vj = (j < 4) ? v0[j] : v1[j - 4];
      mj = beta1 * mj + (1.f - beta1) * gf;
      vj = beta2 * vj + (1.f - beta2) * gf * gf;
      float mhat = mj / bc1;
      float vhat = vj / bc2;
      pf = pf - lr * (mhat / (sqrtf(vhat) + eps) + wd * pf);
      po[j] = f2bf(pf);
      if (j < 4) { m0[j] = mj; v0[j] = vj; }
      else { m1[j - 4] = mj; v1[j - 4] = vj; }
    }
    __builtin_nontemporal_store(po, reinterpret_cast<vec8u*>(p + i));
    __builtin_nontemporal_store(m0, reinterpret_cast<vec4f*>(m + i));
    __builtin_nontemporal_store(m1, reinterpret_cast<vec4f*>(m + i + 4));
    __builtin_nontemporal_store(v0, reinterpret_cast<vec4f*>(v + i));
    __builtin_nontemporal_store(v1, reinterpret_cast<vec4f*>(v + i + 4));
  }
}

// Tail handler for n not divisible by 8 (scalar; runs in the same launch).
__global__ void adamw_tail_kernel(u16* p, const u16* g, float* m, float* v,
                                  long start, long n, float lr, float beta1,
                                  float beta2, float eps, float wd, float bc1,
                                  float bc2, float grad_scale) {
  long i = start + blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  float gf = bf2f(g[i]) * grad_scale;
  float pf = bf2f(p[i]);
  float mj = beta1 * m[i] + (1.f - beta1) * gf;
  float vj = beta2 * v[i] + (1.f - beta2) * gf * gf;
  m[i] = mj; v[i] = vj;
  float mhat = mj / bc1, vhat = vj / bc2;
  p[i] = f2bf(pf - lr * (mhat / (sqrtf(vhat) + eps) + wd * pf));
}

// ---------------------------------------------------------------------------
// Multi-tensor pack/unpack for the GPU data plane (state-dict transfers).
// One kernel moves a whole same-dtype state dict between N scattered
// tensors and one flat buffer (reference does torch.cat + N split copies —
// SURVEY.md §2.9 #8 names this the hand-written-HIP candidate). Segment
// starts in the flat buffer are 16 B-aligned (the host computes padded
// offsets), and torch allocations are 256 B-aligned, so every copy runs as
// uint4 (16 B) vectors with a byte tail. blockIdx.y = segment; x grid-strides
// within it (small segments' blocks exit immediately — launch cost only).
// ---------------------------------------------------------------------------
typedef uint vec4ui __attribute__((ext_vector_type(4)));

__global__ void pack_segments_kernel(const unsigned long long* __restrict__ ptrs,
                                     const long* __restrict__ nbytes,
                                     const long* __restrict__ offs,
                                     char* __restrict__ base, int to_base) {
  const int seg = blockIdx.y;
  const long B = nbytes[seg];
  char* flat = base + offs[seg];                       // 16 B aligned
  char* t = reinterpret_cast<char*>(ptrs[seg]);        // torch alloc: aligned
  char* dst = to_base ? flat : t;
  const char* src = to_base ? (const char*)t : (const char*)flat;
  const long nvec = B >> 4;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    reinterpret_cast<vec4ui*>(dst)[i] =
        reinterpret_cast<const vec4ui*>(src)[i];
  }
  if (blockIdx.x == 0 && threadIdx.x < (B & 15))
    dst[(nvec << 4) + threadIdx.x] = src[(nvec << 4) + threadIdx.x];
}

// ---------------------------------------------------------------------------
// C launchers
// ---------------------------------------------------------------------------
static inline int grid_for(long work, int block, int cap = 2048) {
  long g = (work + block - 1) / block;
  if (g > cap) g = cap;
  if (g < 1) g = 1;
  return (int)g;
}

extern "C" {

void kt_rmsnorm_fwd(const void* x, const void* res, const void* w, void* y,
                    void* s_out, void* invrms, int N, int H, float eps,
                    hipStream_t stream) {
  int grid = N < 2048 ? N : 2048;
  hipLaunchKernelGGL(rmsnorm_fwd_kernel, dim3(grid), dim3(256), 0, stream,
                     (const u16*)x, (const u16*)res, (const u16*)w, (u16*)y,
                     (u16*)s_out, (float*)invrms, N, H, eps);
}

void kt_rmsnorm_bwd(const void* dy, const void* ds, const void* x,
                    const void* w, const void* invrms, void* dx,
                    void* dw_partial, void* dw, int P, int N, int H,
                    hipStream_t stream) {
  int grid = N < P ? N : P;
  size_t lds = (size_t)H * sizeof(float);
  hipLaunchKernelGGL(rmsnorm_bwd_kernel, dim3(grid), dim3(256), lds, stream,
                     (const u16*)dy, (const u16*)ds, (const u16*)x,
                     (const u16*)w, (const float*)invrms, (u16*)dx,
                     (float*)dw_partial, N, H);
  int cgrid = (H + 3) / 4;
  hipLaunchKernelGGL(colsum_bf16_kernel, dim3(cgrid), dim3(256), 0, stream,
                     (const float*)dw_partial, (u16*)dw, grid, H);
}

void kt_rope(const void* x, void* o, const void* cost, const void* sint,
             long total_quads, int S, int Hh, int D, long src_t_stride,
             long src_h_stride, float sign, float oscale,
             hipStream_t stream) {
  int grid = grid_for(total_quads, 256);
  hipLaunchKernelGGL(rope_kernel, dim3(grid), dim3(256), 0, stream,
                     (const u16*)x, (u16*)o, (const float*)cost,
                     (const float*)sint, total_quads, S, Hh, D,
                     src_t_stride, src_h_stride, sign, oscale);
}

void kt_swiglu_fwd(const void* gu, void* out, long N, int I,
                   hipStream_t stream) {
  int grid = grid_for(N * (I >> 3), 256);
  hipLaunchKernelGGL(swiglu_fwd_kernel, dim3(grid), dim3(256), 0, stream,
                     (const u16*)gu, (u16*)out, N, I);
}

void kt_swiglu_bwd(const void* dout, const void* gu, void* dgu, long N, int I,
                   hipStream_t stream) {
  int grid = grid_for(N * (I >> 3), 256);
  hipLaunchKernelGGL(swiglu_bwd_kernel, dim3(grid), dim3(256), 0, stream,
                     (const u16*)dout, (const u16*)gu, (u16*)dgu, N, I);
}

void kt_cross_entropy_fwd(void* logits, const void* targets, void* loss, int N,
                          int V, float scale, long ignore_index,
                          hipStream_t stream) {
  int grid = N < 2048 ? N : 2048;
  hipLaunchKernelGGL(cross_entropy_fwd_kernel, dim3(grid), dim3(256), 0,
                     stream, (u16*)logits, (const long*)targets, (float*)loss,
                     N, V, scale, ignore_index);
}

void kt_adamw(void* p, const void* g, void* m, void* v, long n, float lr,
              float beta1, float beta2, float eps, float wd, float bc1,
              float bc2, float grad_scale, hipStream_t stream) {
  long nvec = n >> 3;
  if (nvec > 0) {
    int grid = grid_for(nvec, 256);
    hipLaunchKernelGGL(adamw_kernel, dim3(grid), dim3(256), 0, stream, (u16*)p,
                       (const u16*)g, (float*)m, (float*)v, n, lr, beta1,
                       beta2, eps, wd, bc1, bc2, grad_scale);
  }
  long tail = n - (nvec << 3);
  if (tail > 0) {
    hipLaunchKernelGGL(adamw_tail_kernel, dim3(1), dim3(64), 0, stream,
                       (u16*)p, (const u16*)g, (float*)m, (float*)v, nvec << 3,
                       n, lr, beta1, beta2, eps, wd, bc1, bc2, grad_scale);
  }
}

void kt_pack_segments(const void* ptrs, const void* nbytes, const void* offs,
                      void* base, int nseg, long max_nbytes, int to_base,
                      hipStream_t stream) {
  int gx = grid_for(max_nbytes >> 4, 256, 1024);
  hipLaunchKernelGGL(pack_segments_kernel, dim3(gx, nseg), dim3(256), 0,
                     stream, (const unsigned long long*)ptrs,
                     (const long*)nbytes, (const long*)offs, (char*)base,
                     to_base);
}

}  // extern "C"
