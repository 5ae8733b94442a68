// Standalone probe: determine the exact lane->element mapping of
// v_mfma_f32_16x16x32_bf16 and v_mfma_f32_32x32x16_bf16 on gfx950.
// Build: hipcc --offload-arch=gfx950 -O2 probe_mfma.hip -o probe_mfma
// Candidates for the K mapping of A/B fragments (8 bf16 per lane):
//   mode 0 ("contig8"): k = (lane>>4)*8 + e
//   mode 1 ("split4"):  k = (lane>>4)*4 + (e&3) + (e>>2)*16
// D mapping assumed (guide): col = lane&15, row = (lane>>4)*4 + reg.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>
#include <cstdlib>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;

__device__ int k_of(int mode, int g, int e) {
  return mode == 0 ? g * 8 + e : g * 4 + (e & 3) + (e >> 2) * 16;
}

// one wave; A (16x32), B (32x16) as fp32 in global; D (16x16) out
__global__ void probe16(const float* A, const float* B, float* D, int amode,
                        int bmode) {
  int l = threadIdx.x;
  int g = l >> 4, i = l & 15;
  bf16x8 a, b;
  for (int e = 0; e < 8; ++e) {
    a[e] = (__bf16)A[i * 32 + k_of(amode, g, e)];
    b[e] = (__bf16)B[k_of(bmode, g, e) * 16 + i];
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  for (int r = 0; r < 4; ++r) D[(g * 4 + r) * 16 + i] = acc[r];
}

// 32x32x16: A (32x16), B (16x32), D (32x32); 4 bf16 per lane for A/B.
//   candidates: k = (lane>>5)*8 + e ("contig8"? only 4 elems) ->
//   mode 0: k = (lane>>5)*4 + e... K=16, lane/32 in {0,1} -> 8 k per group? 16/2=8
//   mode 0: k = (lane>>5)*8 + (e&3) + (e>>2)*?? ; 4 elems: k = (lane>>5)*8 + e? e<4 covers 4
//   mode 0: k = (l>>5)*4 + e        (two groups of 4? covers 8 of 16!?)
//   mode 1: k = (l>>5)*8 + e        (covers e=0..3 -> 8k..; incomplete)
// For 32x32x16 bf16: 64 lanes x 4 elems = 256 = 32*8... A is 32x16=512!? No:
// 32x32x16: A is 32 rows x 16 cols = 512 elems; 64 lanes x 8 elems = 512.
// So A/B fragments are ALSO 8 bf16 (4 VGPRs). acc is 16 f32.
// candidates: mode0 k = (l>>5)*8+e; mode1 k = (l>>5)*4 + (e&3) + (e>>2)*8
__global__ void probe32(const float* A, const float* B, float* D, int amode,
                        int bmode) {
  int l = threadIdx.x;
  int g = l >> 5, i = l & 31;
  bf16x8 a, b;
  for (int e = 0; e < 8; ++e) {
    int k = amode == 0 ? g * 8 + e : g * 4 + (e & 3) + (e >> 2) * 8;
    a[e] = (__bf16)A[i * 16 + k];
    k = bmode == 0 ? g * 8 + e : g * 4 + (e & 3) + (e >> 2) * 8;
    b[e] = (__bf16)B[k * 32 + i];
  }
  f32x16 acc;
  for (int r = 0; r < 16; ++r) acc[r] = 0.f;
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
  // D mapping (guide): col=lane&31, row=(reg&3)+8*(reg>>2)+4*(lane>>5)
  for (int r = 0; r < 16; ++r)
    D[((r & 3) + 8 * (r >> 2) + 4 * g) * 32 + i] = acc[r];
}

int main() {
  const int M16 = 16, K32 = 32;
  float *A, *B, *D;
  hipMallocManaged(&A, 32 * 32 * sizeof(float));
  hipMallocManaged(&B, 32 * 32 * sizeof(float));
  hipMallocManaged(&D, 32 * 32 * sizeof(float));
  srand(7);
  // small ints: bf16-exact, fp32-accum exact
  auto fill = [](float* p, int n) {
    for (int i = 0; i < n; ++i) p[i] = (float)(rand() % 7 - 3);
  };

  // ---- 16x16x32 ----
  fill(A, M16 * K32);
  fill(B, K32 * M16);
  float ref16[16 * 16];
  for (int i = 0; i < 16; ++i)
    for (int j = 0; j < 16; ++j) {
      float s = 0;
      for (int k = 0; k < 32; ++k) s += A[i * 32 + k] * B[k * 16 + j];
      ref16[i * 16 + j] = s;
    }
  for (int am = 0; am < 2; ++am)
    for (int bm = 0; bm < 2; ++bm) {
      hipLaunchKernelGGL(probe16, dim3(1), dim3(64), 0, 0, A, B, D, am, bm);
      hipDeviceSynchronize();
      float maxerr = 0, maxerrT = 0;
      for (int i = 0; i < 16; ++i)
        for (int j = 0; j < 16; ++j) {
          float e = fabsf(D[i * 16 + j] - ref16[i * 16 + j]);
          float eT = fabsf(D[j * 16 + i] - ref16[i * 16 + j]);
          if (e > maxerr) maxerr = e;
          if (eT > maxerrT) maxerrT = eT;
        }
      printf("16x16x32 amode=%d bmode=%d  maxerr=%g  maxerrT=%g\n", am, bm,
             maxerr, maxerrT);
    }

  // ---- 32x32x16 ----
  fill(A, 32 * 16);
  fill(B, 16 * 32);
  static float ref32[32 * 32];
  for (int i = 0; i < 32; ++i)
    for (int j = 0; j < 32; ++j) {
      float s = 0;
      for (int k = 0; k < 16; ++k) s += A[i * 16 + k] * B[k * 32 + j];
      ref32[i * 32 + j] = s;
    }
  for (int am = 0; am < 2; ++am)
    for (int bm = 0; bm < 2; ++bm) {
      hipLaunchKernelGGL(probe32, dim3(1), dim3(64), 0, 0, A, B, D, am, bm);
      hipDeviceSynchronize();
      float maxerr = 0, maxerrT = 0;
      for (int i = 0; i < 32; ++i)
        for (int j = 0; j < 32; ++j) {
          float e = fabsf(D[i * 32 + j] - ref32[i * 32 + j]);
          float eT = fabsf(D[j * 32 + i] - ref32[i * 32 + j]);
          if (e > maxerr) maxerr = e;
          if (eT > maxerrT) maxerrT = eT;
        }
      printf("32x32x16 amode=%d bmode=%d  maxerr=%g  maxerrT=%g\n", am, bm,
             maxerr, maxerrT);
    }
  printf("done\n");
  return 0;
}
