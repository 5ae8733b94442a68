// SSIM on gfx950: 11x11 gaussian-weighted window
// statistics + reduction, matching waternet_amd.utils.metrics._ssim_torch
// (valid convolution, sigma 1.5, k1/k2 configurable).
//
// Input: NCHW fp32 pairs. Each 16x16-thread block computes a 16x16 tile of
// valid window positions for one (n,c) plane from an LDS-staged 26x26 patch;
// block partial sums accumulate into a double scalar.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

#define SSIM_K 11
#define SSIM_HALF 5
#define SSIM_TILE 16
#define SSIM_PATCH (SSIM_TILE + SSIM_K - 1)  // 26

__constant__ float SSIM_W[SSIM_K * SSIM_K];

__global__ void k_ssim(const float* __restrict__ A,
                       const float* __restrict__ B,
                       float* __restrict__ partials, int H, int W, int OH,
                       int OW, float c1, float c2) {
  __shared__ __attribute__((aligned(16))) float sA[SSIM_PATCH][SSIM_PATCH];
  __shared__ __attribute__((aligned(16))) float sB[SSIM_PATCH][SSIM_PATCH];
  __shared__ float red[4];

  const int plane = blockIdx.z;  // n*C + c
  const float* a = A + (long)plane * H * W;
  const float* b = B + (long)plane * H * W;
  const int oy0 = blockIdx.y * SSIM_TILE;
  const int ox0 = blockIdx.x * SSIM_TILE;
  const int tx = threadIdx.x & 15, ty = threadIdx.x >> 4;

  // stage patch (26x26 <= 2 passes of 16x16 + edges)
  for (int yy = ty; yy < SSIM_PATCH; yy += 16)
    for (int xx = tx; xx < SSIM_PATCH; xx += 16) {
      int iy = oy0 + yy, ix = ox0 + xx;
      bool v = iy < H && ix < W;
      sA[yy][xx] = v ? a[(long)iy * W + ix] : 0.f;
      sB[yy][xx] = v ? b[(long)iy * W + ix] : 0.f;
    }
  __syncthreads();

  float ssim = 0.f;
  const int oy = oy0 + ty, ox = ox0 + tx;
  if (oy < OH && ox < OW) {
    float sx = 0.f, sy = 0.f, sxx = 0.f, syy = 0.f, sxy = 0.f;
#pragma unroll 1
    for (int ky = 0; ky < SSIM_K; ++ky) {
#pragma unroll
      for (int kx = 0; kx < SSIM_K; ++kx) {
        float w = SSIM_W[ky * SSIM_K + kx];
        float xa = sA[ty + ky][tx + kx];
        float xb = sB[ty + ky][tx + kx];
        sx += w * xa;
        sy += w * xb;
        sxx += w * xa * xa;
        syy += w * xb * xb;
        sxy += w * xa * xb;
      }
    }
    float vx = sxx - sx * sx;
    float vy = syy - sy * sy;
    float cxy = sxy - sx * sy;
    float num = (2.f * sx * sy + c1) * (2.f * cxy + c2);
    float den = (sx * sx + sy * sy + c1) * (vx + vy + c2);
    ssim = num / den;
  }
  // wave reduce -> LDS -> ONE fp32 partial per block (a single f64
  // accumulator serialized ~9.4K atomics and dominated the kernel)
  for (int off = 32; off > 0; off >>= 1) ssim += __shfl_down(ssim, off, 64);
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = ssim;
  __syncthreads();
  if (threadIdx.x == 0) {
    const int bid =
        (blockIdx.z * gridDim.y + blockIdx.y) * gridDim.x + blockIdx.x;
    partials[bid] = red[0] + red[1] + red[2] + red[3];
  }
}

// NHWC bf16 variant (full-NHWC metric path): identical math; blockIdx.z
// decodes (n, logical channel) and loads convert bf16 -> fp32 at stage time.
__global__ void k_ssim_nhwc(const bf16_t* __restrict__ A,
                            const bf16_t* __restrict__ B,
                            float* __restrict__ partials, int H, int W,
                            int OH, int OW, int Cp, int Clog, float c1,
                            float c2) {
  __shared__ __attribute__((aligned(16))) float sA[SSIM_PATCH][SSIM_PATCH];
  __shared__ __attribute__((aligned(16))) float sB[SSIM_PATCH][SSIM_PATCH];
  __shared__ float red[4];

  const int n = blockIdx.z / Clog, c = blockIdx.z - n * Clog;
  const bf16_t* a = A + (long)n * H * W * Cp + c;
  const bf16_t* b = B + (long)n * H * W * Cp + c;
  const int oy0 = blockIdx.y * SSIM_TILE;
  const int ox0 = blockIdx.x * SSIM_TILE;
  const int tx = threadIdx.x & 15, ty = threadIdx.x >> 4;

  for (int yy = ty; yy < SSIM_PATCH; yy += 16)
    for (int xx = tx; xx < SSIM_PATCH; xx += 16) {
      int iy = oy0 + yy, ix = ox0 + xx;
      bool v = iy < H && ix < W;
      const long off = ((long)iy * W + ix) * Cp;
      sA[yy][xx] = v ? bf2f(a[off]) : 0.f;
      sB[yy][xx] = v ? bf2f(b[off]) : 0.f;
    }
  __syncthreads();

  float ssim = 0.f;
  const int oy = oy0 + ty, ox = ox0 + tx;
  if (oy < OH && ox < OW) {
    float sx = 0.f, sy = 0.f, sxx = 0.f, syy = 0.f, sxy = 0.f;
#pragma unroll 1
    for (int ky = 0; ky < SSIM_K; ++ky) {
#pragma unroll
      for (int kx = 0; kx < SSIM_K; ++kx) {
        float w = SSIM_W[ky * SSIM_K + kx];
        float xa = sA[ty + ky][tx + kx];
        float xb = sB[ty + ky][tx + kx];
        sx += w * xa;
        sy += w * xb;
        sxx += w * xa * xa;
        syy += w * xb * xb;
        sxy += w * xa * xb;
      }
    }
    float vx = sxx - sx * sx;
    float vy = syy - sy * sy;
    float cxy = sxy - sx * sy;
    float num = (2.f * sx * sy + c1) * (2.f * cxy + c2);
    float den = (sx * sx + sy * sy + c1) * (vx + vy + c2);
    ssim = num / den;
  }
  for (int off = 32; off > 0; off >>= 1) ssim += __shfl_down(ssim, off, 64);
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = ssim;
  __syncthreads();
  if (threadIdx.x == 0) {
    const int bid =
        (blockIdx.z * gridDim.y + blockIdx.y) * gridDim.x + blockIdx.x;
    partials[bid] = red[0] + red[1] + red[2] + red[3];
  }
}

static void ssim_init_weights() {
  static bool weights_ready = false;
  if (!weights_ready) {
    float g[SSIM_K], w[SSIM_K * SSIM_K];
    float sum = 0.f;
    for (int i = 0; i < SSIM_K; ++i) {
      float d = i - (SSIM_K - 1) / 2.0f;
      g[i] = expf(-(d * d) / (2.f * 1.5f * 1.5f));
      sum += g[i];
    }
    for (int i = 0; i < SSIM_K; ++i) g[i] /= sum;
    for (int i = 0; i < SSIM_K; ++i)
      for (int j = 0; j < SSIM_K; ++j) w[i * SSIM_K + j] = g[i] * g[j];
    hipMemcpyToSymbol(HIP_SYMBOL(SSIM_W), w, sizeof(w));
    weights_ready = true;
  }
}

at::Tensor ssim_sum_nhwc(const at::Tensor& a, const at::Tensor& b,
                         int64_t Clog, double data_range, double k1,
                         double k2) {
  TORCH_CHECK(a.is_cuda() && a.dtype() == at::kBFloat16 && a.dim() == 4);
  TORCH_CHECK(a.sizes() == b.sizes());
  ssim_init_weights();
  const int N = a.size(0), H = a.size(1), W = a.size(2), Cp = a.size(3);
  const int OH = H - SSIM_K + 1, OW = W - SSIM_K + 1;
  TORCH_CHECK(OH > 0 && OW > 0, "image smaller than SSIM window");
  const float c1 = (float)((k1 * data_range) * (k1 * data_range));
  const float c2 = (float)((k2 * data_range) * (k2 * data_range));
  dim3 grid((OW + 15) / 16, (OH + 15) / 16, N * (int)Clog);
  auto partials = at::empty({(long)grid.x * grid.y * grid.z},
                            a.options().dtype(at::kFloat));
  hipStream_t stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(k_ssim_nhwc, grid, dim3(256), 0, stream,
                     (const bf16_t*)a.contiguous().data_ptr(),
                     (const bf16_t*)b.contiguous().data_ptr(),
                     partials.data_ptr<float>(), H, W, OH, OW, Cp,
                     (int)Clog, c1, c2);
  HIP_CHECK_LAST();
  return partials.sum(at::kDouble);  // caller divides by N*Clog*OH*OW
}

at::Tensor ssim_sum(const at::Tensor& a, const at::Tensor& b,
                    double data_range, double k1, double k2) {
  TORCH_CHECK(a.is_cuda() && a.dtype() == at::kFloat && a.dim() == 4);
  TORCH_CHECK(a.sizes() == b.sizes());
  ssim_init_weights();
  const int N = a.size(0), C = a.size(1), H = a.size(2), W = a.size(3);
  const int OH = H - SSIM_K + 1, OW = W - SSIM_K + 1;
  TORCH_CHECK(OH > 0 && OW > 0, "image smaller than SSIM window");
  const float c1 = (float)((k1 * data_range) * (k1 * data_range));
  const float c2 = (float)((k2 * data_range) * (k2 * data_range));
  dim3 grid((OW + 15) / 16, (OH + 15) / 16, N * C);
  auto partials = at::empty({(long)grid.x * grid.y * grid.z},
                            a.options().dtype(at::kFloat));
  hipStream_t stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(k_ssim, grid, dim3(256), 0, stream,
                     a.contiguous().data_ptr<float>(),
                     b.contiguous().data_ptr<float>(),
                     partials.data_ptr<float>(), H, W, OH, OW, c1, c2);
  HIP_CHECK_LAST();
  return partials.sum(at::kDouble);  // caller divides by N*C*OH*OW
}
