// 2x2 stride-2 max pooling, NHWC bf16 (VGG19's pools — SURVEY §2.2 K17).
// Forward stores a 2-bit argmax index per element for the backward scatter.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {
inline hipStream_t cur_stream() { return at::cuda::getCurrentHIPStream(); }
}

__global__ void k_maxpool_fwd(const bf16_t* __restrict__ x,
                              bf16_t* __restrict__ y,
                              uint8_t* __restrict__ idx, long NOHW, int OH,
                              int OW, int H, int W, int Cp) {
  // one thread per (n, oy, ox, c)
  const long total = NOHW * Cp;
  for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += (long)gridDim.x * blockDim.x) {
    const long p = t / Cp;
    const int c = (int)(t - p * Cp);
    const long n = p / ((long)OH * OW);
    const int rem = (int)(p - n * OH * OW);
    const int oy = rem / OW, ox = rem - (rem / OW) * OW;
    const int iy = oy * 2, ix = ox * 2;
    const long base = ((n * H + iy) * W + ix) * Cp + c;
    float v0 = bf2f(x[base]);
    float v1 = bf2f(x[base + Cp]);
    float v2 = bf2f(x[base + (long)W * Cp]);
    float v3 = bf2f(x[base + (long)W * Cp + Cp]);
    float m = v0;
    int a = 0;
    if (v1 > m) { m = v1; a = 1; }
    if (v2 > m) { m = v2; a = 2; }
    if (v3 > m) { m = v3; a = 3; }
    y[t] = f2bf(m);
    idx[t] = (uint8_t)a;
  }
}

__global__ void k_maxpool_bwd(const bf16_t* __restrict__ dy,
                              const uint8_t* __restrict__ idx,
                              bf16_t* __restrict__ dx, long NOHW, int OH,
                              int OW, int H, int W, int Cp) {
  const long total = NOHW * Cp;
  for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += (long)gridDim.x * blockDim.x) {
    const long p = t / Cp;
    const int c = (int)(t - p * Cp);
    const long n = p / ((long)OH * OW);
    const int rem = (int)(p - n * OH * OW);
    const int oy = rem / OW, ox = rem - (rem / OW) * OW;
    const int a = idx[t];
    const int iy = oy * 2 + (a >> 1), ix = ox * 2 + (a & 1);
    dx[((n * H + iy) * W + ix) * Cp + c] = dy[t];
  }
}

std::vector<at::Tensor> maxpool2x2_fwd(const at::Tensor& x) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == at::kBFloat16 && x.dim() == 4);
  const long N = x.size(0), H = x.size(1), W = x.size(2), Cp = x.size(3);
  const long OH = H / 2, OW = W / 2;
  auto y = at::empty({N, OH, OW, Cp}, x.options());
  auto idx = at::empty({N, OH, OW, Cp}, x.options().dtype(at::kByte));
  const long NOHW = N * OH * OW;
  const int blocks = (int)std::min<long>(4096, (NOHW * Cp + 255) / 256);
  hipLaunchKernelGGL(k_maxpool_fwd, dim3(blocks), dim3(256), 0, cur_stream(),
                     (const bf16_t*)x.data_ptr(), (bf16_t*)y.data_ptr(),
                     idx.data_ptr<uint8_t>(), NOHW, (int)OH, (int)OW, (int)H,
                     (int)W, (int)Cp);
  HIP_CHECK_LAST();
  return {y, idx};
}

at::Tensor maxpool2x2_bwd(const at::Tensor& dy, const at::Tensor& idx,
                          int64_t H, int64_t W) {
  const long N = dy.size(0), OH = dy.size(1), OW = dy.size(2),
             Cp = dy.size(3);
  auto dx = at::zeros({N, H, W, Cp}, dy.options());
  const long NOHW = N * OH * OW;
  const int blocks = (int)std::min<long>(4096, (NOHW * Cp + 255) / 256);
  hipLaunchKernelGGL(k_maxpool_bwd, dim3(blocks), dim3(256), 0, cur_stream(),
                     (const bf16_t*)dy.data_ptr(), idx.data_ptr<uint8_t>(),
                     (bf16_t*)dx.data_ptr(), NOHW, (int)OH, (int)OW, (int)H,
                     (int)W, (int)Cp);
  HIP_CHECK_LAST();
  return dx;
}
