// Empirical layout probe for v_mfma_f32_16x16x32_bf16 (gfx950):
// determines the lane->element mapping of the A and B operands by feeding
// marker values and reading D with the known C/D mapping
// (col = lane&15, row = (lane>>4)*4 + reg). Build standalone:
//   hipcc --offload-arch=gfx950 -O2 probe_mfma.hip -o probe_mfma
#include <hip/hip_runtime.h>
#include <cstdio>

typedef unsigned short u16;
typedef __bf16 bf16;
typedef bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

__device__ __forceinline__ u16 f2bf(float f) {
  union { float f; unsigned int u; } x{f};
  x.u += 0x7fff + ((x.u >> 16) & 1);
  return (u16)(x.u >> 16);
}
__device__ __forceinline__ float bf2f(u16 v) {
  union { unsigned int u; float f; } x{(unsigned int)v << 16};
  return x.f;
}

// hypothesis: A[16x32] lane l holds row=l&15, k=8*(l>>4)+i (i=0..7)
//             B[32x16] lane l holds col=l&15, k=8*(l>>4)+i
// verify by computing D = A @ B with A[i][k]=i+k/100, B=delta(k,j) and
// delta(k,j+16), checking D[i][j] == A[i][j] / A[i][j+16].
__global__ void probe(float* d_out, int b_shift) {
  const int lane = threadIdx.x & 63;
  bf16x8 a, b;
  for (int i = 0; i < 8; ++i) {
    int arow = lane & 15, ak = 8 * (lane >> 4) + i;
    u16 av = f2bf((float)arow + (float)ak / 100.f);
    a[i] = *(bf16*)&av;
    int bcol = lane & 15, bk = 8 * (lane >> 4) + i;
    u16 bv = f2bf((bk == bcol + b_shift) ? 1.f : 0.f);
    b[i] = *(bf16*)&bv;
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  // D mapping: col=lane&15, row=(lane>>4)*4+r
  for (int r = 0; r < 4; ++r) {
    int row = (lane >> 4) * 4 + r, col = lane & 15;
    d_out[row * 16 + col] = acc[r];
  }
}

int main() {
  float* d;
  hipMalloc(&d, 256 * sizeof(float));
  for (int shift : {0, 16}) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d, shift);
    hipDeviceSynchronize();
    float h[256];
    hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    int bad = 0;
    for (int i = 0; i < 16; ++i)
      for (int j = 0; j < 16; ++j) {
        float expect = (float)i + (float)(j + shift) / 100.f;
        // bf16-rounded expectation
        union { float f; unsigned int u; } x{expect};
        x.u = (x.u + 0x7fff + ((x.u >> 16) & 1)) & 0xffff0000u;
        if (h[i * 16 + j] != x.f && ++bad < 4)
          printf("shift%d mismatch D[%d][%d]=%f expect %f\n", shift, i, j,
                 h[i * 16 + j], x.f);
      }
    printf("shift=%d: %s (%d mismatches)\n", shift,
           bad ? "LAYOUT HYPOTHESIS WRONG" : "layout confirmed", bad);
  }
  return 0;
}
