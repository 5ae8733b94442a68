// GPU-native preprocess pipeline: white balance, gamma
// correction and CLAHE hist-eq as CDNA4 kernels, replacing the reference's
// per-image CPU loop (the reference's dataloader bottleneck — data.py:6-90).
//
// Semantics match waternet_amd/data/transforms.py (the CPU reference that
// the unit tests compare against):
//   WB: per-channel exact integer histogram -> np.quantile-style linear
//       interpolation order statistics -> clip + min-max stretch (double
//       precision scale, truncating uint8 cast).
//   Gamma: 256-entry LUT computed host-side in float64 (bit-exact).
//   CLAHE: per-tile 256-bin histograms, integer clip+redistribute, CDF LUT
//       (round-half-even), bilinear LUT interpolation on the LAB L channel.
//   LAB<->RGB: float math, D65/sRGB, OpenCV 8-bit scaling.
//
// Requires H and W divisible by the 8x8 tile grid (112, 512, 1080p all are);
// the Python wrapper falls back to the CPU path otherwise.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {
inline hipStream_t cur_stream() { return at::cuda::getCurrentHIPStream(); }
}

// ---------------------------------------------------------------------------
// Histograms (RGB channels, whole image) for white balance
// ---------------------------------------------------------------------------

__global__ void k_rgb_hist(const uint8_t* __restrict__ img,
                           unsigned int* __restrict__ hist,  // (N,3,256)
                           long HW, int nsplit) {
  __shared__ unsigned int h[3 * 256];
  for (int i = threadIdx.x; i < 3 * 256; i += blockDim.x) h[i] = 0;
  __syncthreads();
  const long n = blockIdx.x;
  const long chunk = (HW + nsplit - 1) / nsplit;
  const long p0 = blockIdx.y * chunk;
  const long p1 = min(p0 + chunk, HW);
  const uint8_t* base = img + n * HW * 3;
  for (long p = p0 + threadIdx.x; p < p1; p += blockDim.x) {
    atomicAdd(&h[0 * 256 + base[p * 3 + 0]], 1u);
    atomicAdd(&h[1 * 256 + base[p * 3 + 1]], 1u);
    atomicAdd(&h[2 * 256 + base[p * 3 + 2]], 1u);
  }
  __syncthreads();
  unsigned int* out = hist + n * 3 * 256;
  for (int i = threadIdx.x; i < 3 * 256; i += blockDim.x)
    if (h[i]) atomicAdd(&out[i], h[i]);
}

// ---------------------------------------------------------------------------
// White-balance parameters: channel sums -> satLevels -> quantiles (exact
// np.quantile linear interpolation over the integer histogram).
// ---------------------------------------------------------------------------

__device__ double quantile_from_hist(const unsigned int* h, long M, double q) {
  // np.quantile (linear): pos = q*(M-1); v = a[floor] + frac*(a[ceil]-a[floor])
  double pos = q * (double)(M - 1);
  long lo = (long)floor(pos);
  double frac = pos - (double)lo;
  long hi = frac > 0.0 ? lo + 1 : lo;
  // order statistics lo, hi via cumulative counts
  long cum = 0;
  int vlo = -1, vhi = -1;
  for (int v = 0; v < 256; ++v) {
    cum += h[v];
    if (vlo < 0 && cum > lo) vlo = v;
    if (vhi < 0 && cum > hi) { vhi = v; break; }
  }
  if (vlo < 0) vlo = 255;
  if (vhi < 0) vhi = 255;
  return (double)vlo + frac * (double)(vhi - vlo);
}

__global__ void k_wb_params(const unsigned int* __restrict__ hist,
                            float* __restrict__ params,  // (N,3,2) lo,hi
                            long HW) {
  const long n = blockIdx.x;
  const int ch = threadIdx.x;  // 3 lanes do work; no early return (barrier)
  __shared__ double sums[3];
  const unsigned int* h = hist + (n * 3 + min(ch, 2)) * 256;
  if (ch < 3) {
    unsigned long long s = 0;
    for (int v = 0; v < 256; ++v) s += (unsigned long long)h[v] * v;
    sums[ch] = (double)s;
  }
  __syncthreads();
  if (ch < 3) {
    double maxsum = fmax(sums[0], fmax(sums[1], sums[2]));
    double ratio = maxsum / sums[ch];
    double sat = 0.005 * ratio;
    // degenerate channel (zero sum -> inf/nan sat, where the reference
    // crashes): fall back to the un-saturated min/max stretch
    if (!(isfinite(sat) && sat <= 0.5)) sat = 0.0;
    double lo = quantile_from_hist(h, HW, sat);
    double hi = quantile_from_hist(h, HW, 1.0 - sat);
    params[(n * 3 + ch) * 2 + 0] = (float)lo;
    params[(n * 3 + ch) * 2 + 1] = (float)hi;
  }
}

// ---------------------------------------------------------------------------
// Apply white balance + gamma LUT in one pass over raw
// ---------------------------------------------------------------------------

__global__ void k_wb_gc_apply(const uint8_t* __restrict__ img,
                              const float* __restrict__ params,
                              const uint8_t* __restrict__ gamma_lut,
                              uint8_t* __restrict__ wb_out,
                              uint8_t* __restrict__ gc_out, long HW) {
  const long n = blockIdx.y;
  const uint8_t* base = img + n * HW * 3;
  uint8_t* wbo = wb_out + n * HW * 3;
  uint8_t* gco = gc_out + n * HW * 3;
  double lo[3], hi[3], scale[3];
#pragma unroll
  for (int c = 0; c < 3; ++c) {
    lo[c] = params[(n * 3 + c) * 2 + 0];
    hi[c] = params[(n * 3 + c) * 2 + 1];
    scale[c] = hi[c] > lo[c] ? 255.0 / (hi[c] - lo[c]) : 0.0;
  }
  for (long p = (long)blockIdx.x * blockDim.x + threadIdx.x; p < HW;
       p += (long)gridDim.x * blockDim.x) {
#pragma unroll
    for (int c = 0; c < 3; ++c) {
      double v = (double)base[p * 3 + c];
      v = v < lo[c] ? lo[c] : (v > hi[c] ? hi[c] : v);
      double r = (v - lo[c]) * scale[c];
      wbo[p * 3 + c] = (uint8_t)r;  // truncating cast, as .astype(np.uint8)
      gco[p * 3 + c] = gamma_lut[base[p * 3 + c]];
    }
  }
}

// ---------------------------------------------------------------------------
// RGB -> LAB (OpenCV 8-bit scaling), float math
// ---------------------------------------------------------------------------

WN_DEVFN float srgb_lin(float s) {
  return s <= 0.04045f ? s / 12.92f : powf((s + 0.055f) / 1.055f, 2.4f);
}
WN_DEVFN float srgb_delin(float l) {
  l = fminf(fmaxf(l, 0.f), 1.f);
  return l <= 0.0031308f ? l * 12.92f : 1.055f * powf(l, 1.f / 2.4f) - 0.055f;
}
WN_DEVFN float lab_f(float t) {
  const float d = 6.f / 29.f;
  return t > d * d * d ? cbrtf(t) : t / (3.f * d * d) + 4.f / 29.f;
}
WN_DEVFN float lab_finv(float ft) {
  const float d = 6.f / 29.f;
  return ft > d ? ft * ft * ft : 3.f * d * d * (ft - 4.f / 29.f);
}

__constant__ float RGB2XYZ[9] = {0.412453f, 0.357580f, 0.180423f,
                                 0.212671f, 0.715160f, 0.072169f,
                                 0.019334f, 0.119193f, 0.950227f};
__constant__ float XYZ2RGB[9] = {3.240479f,  -1.537150f, -0.498535f,
                                 -0.969256f, 1.875992f,  0.041556f,
                                 0.055648f,  -0.204043f, 1.057311f};
__constant__ float WHITE_PT[3] = {0.950456f, 1.0f, 1.088754f};

__global__ void k_rgb2lab(const uint8_t* __restrict__ rgb,
                          uint8_t* __restrict__ lab, long total) {
  for (long p = (long)blockIdx.x * blockDim.x + threadIdx.x; p < total;
       p += (long)gridDim.x * blockDim.x) {
    float r = srgb_lin(rgb[p * 3 + 0] / 255.f);
    float g = srgb_lin(rgb[p * 3 + 1] / 255.f);
    float b = srgb_lin(rgb[p * 3 + 2] / 255.f);
    float X = RGB2XYZ[0] * r + RGB2XYZ[1] * g + RGB2XYZ[2] * b;
    float Y = RGB2XYZ[3] * r + RGB2XYZ[4] * g + RGB2XYZ[5] * b;
    float Z = RGB2XYZ[6] * r + RGB2XYZ[7] * g + RGB2XYZ[8] * b;
    float fx = lab_f(X / WHITE_PT[0]);
    float fy = lab_f(Y / WHITE_PT[1]);
    float fz = lab_f(Z / WHITE_PT[2]);
    float L = 116.f * fy - 16.f;
    float A = 500.f * (fx - fy);
    float B = 200.f * (fy - fz);
    lab[p * 3 + 0] = (uint8_t)fminf(fmaxf(rintf(L * 255.f / 100.f), 0.f), 255.f);
    lab[p * 3 + 1] = (uint8_t)fminf(fmaxf(rintf(A + 128.f), 0.f), 255.f);
    lab[p * 3 + 2] = (uint8_t)fminf(fmaxf(rintf(B + 128.f), 0.f), 255.f);
  }
}

// ---------------------------------------------------------------------------
// CLAHE on the L channel: per-tile histograms -> clipped LUTs -> bilinear
// interpolation fused with LAB->RGB.
// ---------------------------------------------------------------------------

#define TGRID 8

__global__ void k_clahe_hist(const uint8_t* __restrict__ lab,
                             unsigned int* __restrict__ hists,  // (N,64,256)
                             int H, int W) {
  __shared__ unsigned int h[256];
  for (int i = threadIdx.x; i < 256; i += blockDim.x) h[i] = 0;
  __syncthreads();
  const int n = blockIdx.x;
  const int tile = blockIdx.y;
  const int ty = tile / TGRID, tx = tile % TGRID;
  const int th = H / TGRID, tw = W / TGRID;
  const uint8_t* base = lab + (long)n * H * W * 3;
  const int npix = th * tw;
  for (int p = threadIdx.x; p < npix; p += blockDim.x) {
    int yy = ty * th + p / tw;
    int xx = tx * tw + p - (p / tw) * tw;
    atomicAdd(&h[base[((long)yy * W + xx) * 3]], 1u);
  }
  __syncthreads();
  unsigned int* out = hists + ((long)n * 64 + tile) * 256;
  for (int i = threadIdx.x; i < 256; i += blockDim.x) out[i] = h[i];
}

__global__ void k_clahe_lut(unsigned int* __restrict__ hists,  // in-place ok
                            uint8_t* __restrict__ luts,        // (N,64,256)
                            int tileArea, int clip) {
  __shared__ unsigned int h[256];
  __shared__ unsigned int red[256];
  const long tid = (long)blockIdx.x;  // n*64 + tile
  unsigned int* src = hists + tid * 256;
  const int i = threadIdx.x;
  h[i] = src[i];
  __syncthreads();
  // compute excess
  unsigned int e = h[i] > (unsigned)clip ? h[i] - clip : 0;
  red[i] = e;
  __syncthreads();
  for (int s = 128; s > 0; s >>= 1) {
    if (i < s) red[i] += red[i + s];
    __syncthreads();
  }
  const unsigned int excess = red[0];
  __syncthreads();
  unsigned int v = min(h[i], (unsigned)clip);
  v += excess / 256;
  unsigned int residual = excess - (excess / 256) * 256;
  if (residual) {
    unsigned int step = max(256u / residual, 1u);
    if (i % step == 0 && (unsigned)(i / step) < residual) v += 1;
  }
  h[i] = v;
  __syncthreads();
  // inclusive scan (Hillis-Steele over 256)
  for (int s = 1; s < 256; s <<= 1) {
    unsigned int add = (i >= s) ? h[i - s] : 0;
    __syncthreads();
    h[i] += add;
    __syncthreads();
  }
  float lutScale = 255.f / (float)tileArea;
  float r = rintf((float)h[i] * lutScale);
  luts[tid * 256 + i] = (uint8_t)fminf(fmaxf(r, 0.f), 255.f);
}

__global__ void k_clahe_interp_lab2rgb(const uint8_t* __restrict__ lab,
                                       const uint8_t* __restrict__ luts,
                                       uint8_t* __restrict__ he_rgb, int H,
                                       int W) {
  const int n = blockIdx.y;
  const long HW = (long)H * W;
  const uint8_t* lbase = lab + n * HW * 3;
  const uint8_t* lut = luts + (long)n * 64 * 256;
  uint8_t* out = he_rgb + n * HW * 3;
  const int th = H / TGRID, tw = W / TGRID;
  for (long p = (long)blockIdx.x * blockDim.x + threadIdx.x; p < HW;
       p += (long)gridDim.x * blockDim.x) {
    const int y = (int)(p / W), x = (int)(p - (long)(p / W) * W);
    const uint8_t Lv = lbase[p * 3 + 0];
    float tyf = (float)y / th - 0.5f;
    float txf = (float)x / tw - 0.5f;
    int ty1 = (int)floorf(tyf), tx1 = (int)floorf(txf);
    float ya = tyf - ty1, xa = txf - tx1;
    int ty2 = min(ty1 + 1, TGRID - 1), tx2 = min(tx1 + 1, TGRID - 1);
    ty1 = max(ty1, 0); tx1 = max(tx1, 0);
    float l11 = lut[(ty1 * TGRID + tx1) * 256 + Lv];
    float l12 = lut[(ty1 * TGRID + tx2) * 256 + Lv];
    float l21 = lut[(ty2 * TGRID + tx1) * 256 + Lv];
    float l22 = lut[(ty2 * TGRID + tx2) * 256 + Lv];
    float res = l11 * (1 - xa) * (1 - ya) + l12 * xa * (1 - ya) +
                l21 * (1 - xa) * ya + l22 * xa * ya;
    float newL = fminf(fmaxf(rintf(res), 0.f), 255.f);
    // LAB -> RGB
    float L = newL * 100.f / 255.f;
    float A = (float)lbase[p * 3 + 1] - 128.f;
    float B = (float)lbase[p * 3 + 2] - 128.f;
    float fy = (L + 16.f) / 116.f;
    float fx = fy + A / 500.f;
    float fz = fy - B / 200.f;
    float X = lab_finv(fx) * WHITE_PT[0];
    float Y = lab_finv(fy) * WHITE_PT[1];
    float Z = lab_finv(fz) * WHITE_PT[2];
    float r = srgb_delin(XYZ2RGB[0] * X + XYZ2RGB[1] * Y + XYZ2RGB[2] * Z);
    float g = srgb_delin(XYZ2RGB[3] * X + XYZ2RGB[4] * Y + XYZ2RGB[5] * Z);
    float b = srgb_delin(XYZ2RGB[6] * X + XYZ2RGB[7] * Y + XYZ2RGB[8] * Z);
    out[p * 3 + 0] = (uint8_t)fminf(fmaxf(rintf(r * 255.f), 0.f), 255.f);
    out[p * 3 + 1] = (uint8_t)fminf(fmaxf(rintf(g * 255.f), 0.f), 255.f);
    out[p * 3 + 2] = (uint8_t)fminf(fmaxf(rintf(b * 255.f), 0.f), 255.f);
  }
}

// ---------------------------------------------------------------------------
// Host orchestration
// ---------------------------------------------------------------------------

std::vector<at::Tensor> preprocess_all(const at::Tensor& raw_u8) {
  TORCH_CHECK(raw_u8.is_cuda() && raw_u8.dtype() == at::kByte &&
              raw_u8.dim() == 4 && raw_u8.size(3) == 3,
              "raw must be (N,H,W,3) uint8 CUDA");
  const int N = raw_u8.size(0), H = raw_u8.size(1), W = raw_u8.size(2);
  TORCH_CHECK(H % TGRID == 0 && W % TGRID == 0,
              "GPU CLAHE requires H,W divisible by 8 (got ", H, "x", W, ")");
  const long HW = (long)H * W;
  auto raw = raw_u8.contiguous();
  auto opts = raw.options();
  auto stream = cur_stream();

  auto wb = at::empty_like(raw);
  auto gc = at::empty_like(raw);
  auto he = at::empty_like(raw);
  auto hist = at::zeros({N, 3, 256}, opts.dtype(at::kInt));
  auto params = at::empty({N, 3, 2}, opts.dtype(at::kFloat));
  auto lab = at::empty_like(raw);
  auto thists = at::empty({N, 64, 256}, opts.dtype(at::kInt));
  auto luts = at::empty({N, 64, 256}, opts.dtype(at::kByte));

  // gamma LUT: float64-exact, computed once and cached on-device (keeps the
  // kernel sequence hipGraph-capturable: no per-call pageable H2D copy)
  static at::Tensor glut;
  if (!glut.defined() || glut.device() != raw.device()) {
    uint8_t glut_host[256];
    for (int v = 0; v < 256; ++v) {
      double g = pow((double)v / 255.0, 0.7) * 255.0;
      g = g < 0 ? 0 : (g > 255 ? 255 : g);
      glut_host[v] = (uint8_t)g;
    }
    glut = at::from_blob(glut_host, {256}, at::kByte).to(raw.device());
  }

  const int nsplit = std::max(1, std::min(16, (int)(HW / 65536)));
  hipLaunchKernelGGL(k_rgb_hist, dim3(N, nsplit), dim3(256), 0, stream,
                     raw.data_ptr<uint8_t>(),
                     (unsigned int*)hist.data_ptr<int>(), HW, nsplit);
  hipLaunchKernelGGL(k_wb_params, dim3(N), dim3(64), 0, stream,
                     (const unsigned int*)hist.data_ptr<int>(),
                     params.data_ptr<float>(), HW);
  hipLaunchKernelGGL(k_wb_gc_apply,
                     dim3(std::min<long>(1024, (HW + 255) / 256), N),
                     dim3(256), 0, stream, raw.data_ptr<uint8_t>(),
                     params.data_ptr<float>(), glut.data_ptr<uint8_t>(),
                     wb.data_ptr<uint8_t>(), gc.data_ptr<uint8_t>(), HW);
  const long total = (long)N * HW;
  hipLaunchKernelGGL(k_rgb2lab,
                     dim3(std::min<long>(4096, (total + 255) / 256)),
                     dim3(256), 0, stream, raw.data_ptr<uint8_t>(),
                     lab.data_ptr<uint8_t>(), total);
  hipLaunchKernelGGL(k_clahe_hist, dim3(N, 64), dim3(256), 0, stream,
                     lab.data_ptr<uint8_t>(),
                     (unsigned int*)thists.data_ptr<int>(), H, W);
  const int tileArea = (H / TGRID) * (W / TGRID);
  const int clip = std::max((int)(0.1 * tileArea / 256.0), 1);
  hipLaunchKernelGGL(k_clahe_lut, dim3(N * 64), dim3(256), 0, stream,
                     (unsigned int*)thists.data_ptr<int>(),
                     luts.data_ptr<uint8_t>(), tileArea, clip);
  hipLaunchKernelGGL(k_clahe_interp_lab2rgb,
                     dim3(std::min<long>(1024, (HW + 255) / 256), N),
                     dim3(256), 0, stream, lab.data_ptr<uint8_t>(),
                     luts.data_ptr<uint8_t>(), he.data_ptr<uint8_t>(), H, W);
  HIP_CHECK_LAST();
  return {wb, gc, he};
}
