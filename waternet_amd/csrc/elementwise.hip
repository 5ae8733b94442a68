// Fused elementwise / layout / reduction kernels for the WaterNet engine
// (gfx950). Covers SURVEY §2.2 K1/K10/K11 (cat/split folding via the input
// builders), K15 (gated fusion fwd/bwd), K16 (ImageNet normalize), K18/K19
// (fused squared-diff reductions), K23/K25 (fused flat Adam), plus the
// NCHW fp32 <-> NHWC bf16 bridges and activation backward.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {
inline hipStream_t cur_stream() { return at::cuda::getCurrentHIPStream(); }
constexpr int TPB = 256;
inline int grid1d(long n, int tpb = TPB, int cap = 4096) {
  return (int)std::min<long>(cap, (n + tpb - 1) / tpb);
}
}  // namespace

// ---------------------------------------------------------------------------
// Input builders: 4x NCHW fp32 (3ch) -> cmg input (N,H,W,16) and three
// refiner inputs (N,H,W,16), bf16, /1 scale (inputs already in [0,1]).
// Folds the reference's torch.cat calls (net.py:46, net.py:76) into one pass.
// ---------------------------------------------------------------------------

__global__ void k_build_inputs(const float* __restrict__ raw,
                               const float* __restrict__ wb,
                               const float* __restrict__ ce,
                               const float* __restrict__ gc,
                               bf16_t* __restrict__ cmg_in,
                               bf16_t* __restrict__ rwb_in,
                               bf16_t* __restrict__ rce_in,
                               bf16_t* __restrict__ rgc_in, long NHW,
                               long HW) {
  for (long p = (long)blockIdx.x * blockDim.x + threadIdx.x; p < NHW;
       p += (long)gridDim.x * blockDim.x) {
    const long n = p / HW, rem = p - n * HW;
    const long base = n * 3 * HW + rem;
    float r0 = raw[base], r1 = raw[base + HW], r2 = raw[base + 2 * HW];
    float w0 = wb[base], w1 = wb[base + HW], w2 = wb[base + 2 * HW];
    float c0 = ce[base], c1 = ce[base + HW], c2 = ce[base + 2 * HW];
    float g0 = gc[base], g1 = gc[base + HW], g2 = gc[base + 2 * HW];
    bf16_t* o = cmg_in + p * 16;
    o[0] = f2bf(r0); o[1] = f2bf(r1); o[2] = f2bf(r2);
    o[3] = f2bf(w0); o[4] = f2bf(w1); o[5] = f2bf(w2);
    o[6] = f2bf(c0); o[7] = f2bf(c1); o[8] = f2bf(c2);
    o[9] = f2bf(g0); o[10] = f2bf(g1); o[11] = f2bf(g2);
    o[12] = o[13] = o[14] = o[15] = f2bf(0.f);
    auto emit = [&](bf16_t* dst, float a0, float a1, float a2) {
      bf16_t* q = dst + p * 16;
      q[0] = f2bf(r0); q[1] = f2bf(r1); q[2] = f2bf(r2);
      q[3] = f2bf(a0); q[4] = f2bf(a1); q[5] = f2bf(a2);
#pragma unroll
      for (int i = 6; i < 16; ++i) q[i] = f2bf(0.f);
    };
    emit(rwb_in, w0, w1, w2);
    emit(rce_in, c0, c1, c2);
    emit(rgc_in, g0, g1, g2);
  }
}

// uint8 variant: builds the same four conv inputs straight from the uint8
// HWC batch tensors (raw + the GPU-preprocess outputs), /255 — removes the
// per-tensor u8->NCHW-fp32 bridges and the NCHW intermediates entirely
// (full-NHWC training path; reference arr2ten training_utils.py:11-27 +
// torch.cat net.py:46/76 in ONE kernel).
__global__ void k_build_inputs_u8(const uint8_t* __restrict__ raw,
                                  const uint8_t* __restrict__ wb,
                                  const uint8_t* __restrict__ ce,
                                  const uint8_t* __restrict__ gc,
                                  bf16_t* __restrict__ cmg_in,
                                  bf16_t* __restrict__ rwb_in,
                                  bf16_t* __restrict__ rce_in,
                                  bf16_t* __restrict__ rgc_in, long NHW) {
  constexpr float S = 1.f / 255.f;
  for (long p = (long)blockIdx.x * blockDim.x + threadIdx.x; p < NHW;
       p += (long)gridDim.x * blockDim.x) {
    const long s = p * 3;
    float r0 = raw[s] * S, r1 = raw[s + 1] * S, r2 = raw[s + 2] * S;
    float w0 = wb[s] * S, w1 = wb[s + 1] * S, w2 = wb[s + 2] * S;
    float c0 = ce[s] * S, c1 = ce[s + 1] * S, c2 = ce[s + 2] * S;
    float g0 = gc[s] * S, g1 = gc[s + 1] * S, g2 = gc[s + 2] * S;
    bf16_t* o = cmg_in + p * 16;
    o[0] = f2bf(r0); o[1] = f2bf(r1); o[2] = f2bf(r2);
    o[3] = f2bf(w0); o[4] = f2bf(w1); o[5] = f2bf(w2);
    o[6] = f2bf(c0); o[7] = f2bf(c1); o[8] = f2bf(c2);
    o[9] = f2bf(g0); o[10] = f2bf(g1); o[11] = f2bf(g2);
    o[12] = o[13] = o[14] = o[15] = f2bf(0.f);
    auto emit = [&](bf16_t* dst, float a0, float a1, float a2) {
      bf16_t* q = dst + p * 16;
      q[0] = f2bf(r0); q[1] = f2bf(r1); q[2] = f2bf(r2);
      q[3] = f2bf(a0); q[4] = f2bf(a1); q[5] = f2bf(a2);
#pragma unroll
      for (int i = 6; i < 16; ++i) q[i] = f2bf(0.f);
    };
    emit(rwb_in, w0, w1, w2);
    emit(rce_in, c0, c1, c2);
    emit(rgc_in, g0, g1, g2);
  }
}

// uint8 HWC -> (N,H,W,Cp) bf16 [0,1] (the ref tensor for the NHWC loss path)
__global__ void k_u8_to_nhwc(const uint8_t* __restrict__ x,
                             bf16_t* __restrict__ y, long NHW, int Cp) {
  constexpr float S = 1.f / 255.f;
  for (long p = (long)blockIdx.x * blockDim.x + threadIdx.x; p < NHW;
       p += (long)gridDim.x * blockDim.x) {
    const uint8_t* s = x + p * 3;
    bf16_t* d = y + (long)p * Cp;
    d[0] = f2bf(s[0] * S);
    d[1] = f2bf(s[1] * S);
    d[2] = f2bf(s[2] * S);
    for (int c = 3; c < Cp; ++c) d[c] = f2bf(0.f);
  }
}

// ---------------------------------------------------------------------------
// NCHW fp32 <-> NHWC bf16 bridges
// ---------------------------------------------------------------------------

// LDS-tiled layout bridges: 256 pixels x 16 channels per block so BOTH
// sides are coalesced (the naive per-pixel loop read NCHW at stride HW and
// wrote 2 B scattered — measured 125 GB/s; these run at HBM rate).
__global__ void k_nchw2nhwc(const float* __restrict__ x,
                            bf16_t* __restrict__ y, long NHW, long HW, int C,
                            int Cp) {
  __shared__ float tile[16][257];
  const int tid = threadIdx.x;
  const long p = (long)blockIdx.x * 256 + tid;
  const int cc0 = blockIdx.z * 16;
  const bool pv = p < NHW;
  const long n = pv ? p / HW : 0;
  const long rem = p - n * HW;
#pragma unroll
  for (int i = 0; i < 16; ++i) {
    const int c = cc0 + i;
    tile[i][tid] =
        (pv && c < C) ? x[(n * C + c) * HW + rem] : 0.f;
  }
  __syncthreads();
  if (pv) {
#pragma unroll
    for (int v8 = 0; v8 < 2; ++v8) {
      bf16x8 v;
#pragma unroll
      for (int e = 0; e < 8; ++e) v[e] = f2bf(tile[v8 * 8 + e][tid]);
      *reinterpret_cast<bf16x8*>(y + p * Cp + cc0 + v8 * 8) = v;
    }
  }
}

__global__ void k_nhwc2nchw(const bf16_t* __restrict__ x,
                            float* __restrict__ y, long NHW, long HW, int C,
                            int Cp) {
  __shared__ float tile[16][257];
  const int tid = threadIdx.x;
  const long p = (long)blockIdx.x * 256 + tid;
  const int cc0 = blockIdx.z * 16;
  const bool pv = p < NHW;
  if (pv) {
#pragma unroll
    for (int v8 = 0; v8 < 2; ++v8) {
      const bf16x8 v =
          *reinterpret_cast<const bf16x8*>(x + p * Cp + cc0 + v8 * 8);
#pragma unroll
      for (int e = 0; e < 8; ++e) tile[v8 * 8 + e][tid] = bf2f(v[e]);
    }
  }
  __syncthreads();
  const long n = pv ? p / HW : 0;
  const long rem = p - n * HW;
#pragma unroll
  for (int i = 0; i < 16; ++i) {
    const int c = cc0 + i;
    if (pv && c < C) y[(n * C + c) * HW + rem] = tile[i][tid];
  }
}

// ---------------------------------------------------------------------------
// Gated fusion (net.py:104-108): out_c = sum_i refined_i_c * map_i
// maps = (N,H,W,16) logical 3 (cmg conv8 sigmoid output)
// refined_i = (N,H,W,16) logical 3
// ---------------------------------------------------------------------------

__global__ void k_fusion_fwd(const bf16_t* __restrict__ maps,
                             const bf16_t* __restrict__ rwb,
                             const bf16_t* __restrict__ rce,
                             const bf16_t* __restrict__ rgc,
                             bf16_t* __restrict__ out, long NHW) {
  for (long p = (long)blockIdx.x * blockDim.x + threadIdx.x; p < NHW;
       p += (long)gridDim.x * blockDim.x) {
    const long o = p * 16;
    float m0 = bf2f(maps[o]), m1 = bf2f(maps[o + 1]), m2 = bf2f(maps[o + 2]);
    bf16_t* dst = out + o;
#pragma unroll
    for (int c = 0; c < 3; ++c) {
      float v = bf2f(rwb[o + c]) * m0 + bf2f(rce[o + c]) * m1 +
                bf2f(rgc[o + c]) * m2;
      dst[c] = f2bf(v);
    }
#pragma unroll
    for (int c = 3; c < 16; ++c) dst[c] = f2bf(0.f);
  }
}

__global__ void k_fusion_bwd(const bf16_t* __restrict__ dout,
                             const bf16_t* __restrict__ maps,
                             const bf16_t* __restrict__ rwb,
                             const bf16_t* __restrict__ rce,
                             const bf16_t* __restrict__ rgc,
                             bf16_t* __restrict__ dmaps,
                             bf16_t* __restrict__ drwb,
                             bf16_t* __restrict__ drce,
                             bf16_t* __restrict__ drgc, long NHW) {
  for (long p = (long)blockIdx.x * blockDim.x + threadIdx.x; p < NHW;
       p += (long)gridDim.x * blockDim.x) {
    const long o = p * 16;
    float m0 = bf2f(maps[o]), m1 = bf2f(maps[o + 1]), m2 = bf2f(maps[o + 2]);
    float dm0 = 0.f, dm1 = 0.f, dm2 = 0.f;
#pragma unroll
    for (int c = 0; c < 3; ++c) {
      float g = bf2f(dout[o + c]);
      dm0 += g * bf2f(rwb[o + c]);
      dm1 += g * bf2f(rce[o + c]);
      dm2 += g * bf2f(rgc[o + c]);
      drwb[o + c] = f2bf(g * m0);
      drce[o + c] = f2bf(g * m1);
      drgc[o + c] = f2bf(g * m2);
    }
    dmaps[o] = f2bf(dm0);
    dmaps[o + 1] = f2bf(dm1);
    dmaps[o + 2] = f2bf(dm2);
#pragma unroll
    for (int c = 3; c < 16; ++c) {
      dmaps[o + c] = f2bf(0.f);
      drwb[o + c] = f2bf(0.f);
      drce[o + c] = f2bf(0.f);
      drgc[o + c] = f2bf(0.f);
    }
  }
}

// ---------------------------------------------------------------------------
// Activation backward: dpre = dY * act'(Y)  (Y = post-activation output)
// ---------------------------------------------------------------------------

__global__ void k_act_bwd(const bf16_t* __restrict__ dy,
                          const bf16_t* __restrict__ y,
                          bf16_t* __restrict__ dpre, long n, int act) {
  // bf16x8 vector path (n is always a multiple of 8 here: NHWC with Cp a
  // multiple of 16); scalar tail kept for generality
  const long n8 = n / 8;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (long)gridDim.x * blockDim.x) {
    const bf16x8 gv = *reinterpret_cast<const bf16x8*>(dy + i * 8);
    const bf16x8 vv = *reinterpret_cast<const bf16x8*>(y + i * 8);
    bf16x8 r;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const float g = bf2f(gv[e]), v = bf2f(vv[e]);
      r[e] = f2bf((act == ACT_RELU) ? (v > 0.f ? g : 0.f)
                                    : g * v * (1.f - v));
    }
    *reinterpret_cast<bf16x8*>(dpre + i * 8) = r;
  }
  for (long i = n8 * 8 + (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    const float g = bf2f(dy[i]), v = bf2f(y[i]);
    dpre[i] = f2bf((act == ACT_RELU) ? (v > 0.f ? g : 0.f)
                                     : g * v * (1.f - v));
  }
}

// act_bwd fused with the bias-gradient column sums (K25 epilogue fusion):
// dpre = dY * act'(Y) while accumulating sum_m dpre[m][k] per block into a
// partial slab — saves the separate bias_grad kernel's full re-read of
// dpre (the WaterNet layers' bias grads; the frozen VGG keeps plain
// act_bwd). Thread->channel-octet mapping is STABLE across grid-stride
// iterations because (gridDim*blockDim*8) % Kp == 0 (host guarantees
// blocks*2048 divisible by Kp; Kp is a power of two <= 128).
__global__ void k_act_bwd_bias(const bf16_t* __restrict__ dy,
                               const bf16_t* __restrict__ y,
                               bf16_t* __restrict__ dpre,
                               float* __restrict__ part,  // [grid][Kp]
                               long n8, int Kp, int act) {
  __shared__ float red[128];  // Kp <= 128
  if (threadIdx.x < Kp) red[threadIdx.x] = 0.f;
  __syncthreads();
  float s[8];
#pragma unroll
  for (int e = 0; e < 8; ++e) s[e] = 0.f;
  const long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (long i = i0; i < n8; i += (long)gridDim.x * blockDim.x) {
    const bf16x8 gv = *reinterpret_cast<const bf16x8*>(dy + i * 8);
    const bf16x8 vv = *reinterpret_cast<const bf16x8*>(y + i * 8);
    bf16x8 r;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const float g = bf2f(gv[e]), v = bf2f(vv[e]);
      const float d = (act == ACT_RELU) ? (v > 0.f ? g : 0.f)
                                        : g * v * (1.f - v);
      r[e] = f2bf(d);
      s[e] += d;
    }
    *reinterpret_cast<bf16x8*>(dpre + i * 8) = r;
  }
  const int c0 = (int)((i0 * 8) % Kp);
#pragma unroll
  for (int e = 0; e < 8; ++e) atomicAdd(&red[c0 + e], s[e]);
  __syncthreads();
  if (threadIdx.x < Kp) part[(long)blockIdx.x * Kp + threadIdx.x] =
      red[threadIdx.x];
}

// slab reduce into db (ACCUMULATES — the arena view may already hold
// autograd-accumulated values)
__global__ void k_abb_reduce(const float* __restrict__ part,
                             float* __restrict__ db, int Kp, int K,
                             int nblk) {
  const int k = blockIdx.x;
  if (k >= K) return;
  __shared__ float red[256];
  float s = 0.f;
  for (int r = threadIdx.x; r < nblk; r += 256)
    s += part[(long)r * Kp + k];
  red[threadIdx.x] = s;
  __syncthreads();
  for (int w = 128; w > 0; w >>= 1) {
    if (threadIdx.x < w) red[threadIdx.x] += red[threadIdx.x + w];
    __syncthreads();
  }
  if (threadIdx.x == 0) db[k] += red[0];
}

// ---------------------------------------------------------------------------
// ImageNet normalize fused with NCHW fp32 -> NHWC bf16 (and backward)
// (train.py:111-116)
// ---------------------------------------------------------------------------

__constant__ float IMNET_MEAN[3] = {0.485f, 0.456f, 0.406f};
__constant__ float IMNET_STD[3] = {0.229f, 0.224f, 0.225f};

__global__ void k_normalize_vgg_fwd(const float* __restrict__ x,
                                    bf16_t* __restrict__ y, long NHW,
                                    long HW, int Cp) {
  for (long p = (long)blockIdx.x * blockDim.x + threadIdx.x; p < NHW;
       p += (long)gridDim.x * blockDim.x) {
    const long n = p / HW, rem = p - n * HW;
    const float* src = x + (n * 3) * HW + rem;
    bf16_t* dst = y + p * Cp;
#pragma unroll
    for (int c = 0; c < 3; ++c)
      dst[c] = f2bf((src[(long)c * HW] - IMNET_MEAN[c]) / IMNET_STD[c]);
    for (int c = 3; c < Cp; ++c) dst[c] = f2bf(0.f);
  }
}

__global__ void k_normalize_vgg_bwd(const bf16_t* __restrict__ dy,
                                    float* __restrict__ dx, long NHW,
                                    long HW, int Cp) {
  for (long p = (long)blockIdx.x * blockDim.x + threadIdx.x; p < NHW;
       p += (long)gridDim.x * blockDim.x) {
    const long n = p / HW, rem = p - n * HW;
    const bf16_t* src = dy + p * Cp;
    float* dst = dx + (n * 3) * HW + rem;
#pragma unroll
    for (int c = 0; c < 3; ++c)
      dst[(long)c * HW] = bf2f(src[c]) / IMNET_STD[c];
  }
}

// NHWC bf16 -> NHWC bf16 in-layout normalize (full-NHWC loss path: the
// WaterNet output and ref stay NHWC from the fusion kernel through the VGG
// towers — no NCHW round trip).
__global__ void k_normalize_nhwc_fwd(const bf16_t* __restrict__ x,
                                     bf16_t* __restrict__ y, long NHW,
                                     int Cp) {
  for (long p = (long)blockIdx.x * blockDim.x + threadIdx.x; p < NHW;
       p += (long)gridDim.x * blockDim.x) {
    const bf16_t* s = x + (long)p * Cp;
    bf16_t* d = y + (long)p * Cp;
#pragma unroll
    for (int c = 0; c < 3; ++c)
      d[c] = f2bf((bf2f(s[c]) - IMNET_MEAN[c]) / IMNET_STD[c]);
    for (int c = 3; c < Cp; ++c) d[c] = f2bf(0.f);
  }
}

__global__ void k_normalize_nhwc_bwd(const bf16_t* __restrict__ dy,
                                     bf16_t* __restrict__ dx, long NHW,
                                     int Cp) {
  for (long p = (long)blockIdx.x * blockDim.x + threadIdx.x; p < NHW;
       p += (long)gridDim.x * blockDim.x) {
    const bf16_t* s = dy + (long)p * Cp;
    bf16_t* d = dx + (long)p * Cp;
#pragma unroll
    for (int c = 0; c < 3; ++c) d[c] = f2bf(bf2f(s[c]) / IMNET_STD[c]);
    for (int c = 3; c < Cp; ++c) d[c] = f2bf(0.f);
  }
}

// ---------------------------------------------------------------------------
// Scaled squared-difference reduction (K18/K19):
//   sum over logical elements of (255*(a-b))^2, fp64 accumulator.
// flat variant (Clog == Cp) and channel-masked variant.
// ---------------------------------------------------------------------------

__global__ void k_sqdiff255_flat(const bf16_t* __restrict__ a,
                                 const bf16_t* __restrict__ b,
                                 double* __restrict__ out, long n) {
  float s = 0.f;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    float d = 255.f * (bf2f(a[i]) - bf2f(b[i]));
    s += d * d;
  }
  // wave reduce then one atomic per wave
  for (int off = 32; off > 0; off >>= 1) s += __shfl_down(s, off, 64);
  if ((threadIdx.x & 63) == 0) atomicAdd(out, (double)s);
}

__global__ void k_sqdiff255_pix(const bf16_t* __restrict__ a,
                                const bf16_t* __restrict__ b,
                                double* __restrict__ out, long NHW, int Clog,
                                int Cp) {
  float s = 0.f;
  for (long p = (long)blockIdx.x * blockDim.x + threadIdx.x; p < NHW;
       p += (long)gridDim.x * blockDim.x) {
    const long o = p * Cp;
    for (int c = 0; c < Clog; ++c) {
      float d = 255.f * (bf2f(a[o + c]) - bf2f(b[o + c]));
      s += d * d;
    }
  }
  for (int off = 32; off > 0; off >>= 1) s += __shfl_down(s, off, 64);
  if ((threadIdx.x & 63) == 0) atomicAdd(out, (double)s);
}

// backward: da = gscale * (a - b), with gscale = grad * 2*255^2/numel
__global__ void k_sqdiff255_bwd(const bf16_t* __restrict__ a,
                                const bf16_t* __restrict__ b,
                                const float* __restrict__ gscale,
                                bf16_t* __restrict__ da, long n, float sign) {
  const float gs = gscale[0] * sign;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    da[i] = f2bf(gs * (bf2f(a[i]) - bf2f(b[i])));
  }
}

// ---------------------------------------------------------------------------
// Fused flat Adam (K23): torch.optim.Adam semantics, fp32 master params.
// ---------------------------------------------------------------------------

// Graph-safe: the step counter and lr live in DEVICE buffers so a
// hipGraph-captured step keeps correct bias correction and per-minibatch
// StepLR semantics under replay (k_adam_tick runs first on the stream).
__global__ void k_adam_tick(int* __restrict__ step) {
  if (threadIdx.x == 0 && blockIdx.x == 0) step[0] += 1;
}

__global__ void k_adam(float* __restrict__ p, const float* __restrict__ g,
                       float* __restrict__ m, float* __restrict__ v, long n,
                       const float* __restrict__ lr_buf, float b1, float b2,
                       float eps, const int* __restrict__ step) {
  const float lr = lr_buf[0];
  const float t = (float)step[0];
  const float bc1 = 1.f - powf(b1, t);
  const float bc2 = 1.f - powf(b2, t);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    float gi = g[i];
    float mi = b1 * m[i] + (1.f - b1) * gi;
    float vi = b2 * v[i] + (1.f - b2) * gi * gi;
    m[i] = mi;
    v[i] = vi;
    float mhat = mi / bc1;
    float vhat = vi / bc2;
    p[i] -= lr * mhat / (sqrtf(vhat) + eps);
  }
}

// ---------------------------------------------------------------------------
// Fused postprocess (K27): NHWC bf16 model output -> uint8 RGB
// (clip to [0,1], *255, truncate — ten2arr semantics training_utils.py:27-43)
// and uint8 NHWC -> NCHW fp32 [0,1] input bridge.
// ---------------------------------------------------------------------------

__global__ void k_out_to_u8(const bf16_t* __restrict__ x,
                            uint8_t* __restrict__ y, long NHW, int Cp) {
  for (long p = (long)blockIdx.x * blockDim.x + threadIdx.x; p < NHW;
       p += (long)gridDim.x * blockDim.x) {
#pragma unroll
    for (int c = 0; c < 3; ++c) {
      float v = bf2f(x[p * Cp + c]);
      v = fminf(fmaxf(v, 0.f), 1.f) * 255.f;
      y[p * 3 + c] = (uint8_t)v;
    }
  }
}

__global__ void k_u8_to_nchw(const uint8_t* __restrict__ x,
                             float* __restrict__ y, long NHW, long HW) {
  for (long p = (long)blockIdx.x * blockDim.x + threadIdx.x; p < NHW;
       p += (long)gridDim.x * blockDim.x) {
    const long n = p / HW, rem = p - n * HW;
    float* dst = y + (n * 3) * HW + rem;
#pragma unroll
    for (int c = 0; c < 3; ++c)
      dst[(long)c * HW] = (float)x[p * 3 + c] * (1.f / 255.f);
  }
}

// ===========================================================================
// Host wrappers
// ===========================================================================

std::vector<at::Tensor> build_inputs(const at::Tensor& raw,
                                     const at::Tensor& wb,
                                     const at::Tensor& ce,
                                     const at::Tensor& gc) {
  TORCH_CHECK(raw.is_cuda() && raw.dtype() == at::kFloat && raw.dim() == 4 &&
              raw.size(1) == 3, "raw must be (N,3,H,W) fp32 CUDA");
  const long N = raw.size(0), H = raw.size(2), W = raw.size(3);
  const long NHW = N * H * W, HW = H * W;
  auto opts = raw.options().dtype(at::kBFloat16);
  auto cmg_in = at::empty({N, H, W, 16}, opts);
  auto rwb = at::empty({N, H, W, 16}, opts);
  auto rce = at::empty({N, H, W, 16}, opts);
  auto rgc = at::empty({N, H, W, 16}, opts);
  hipLaunchKernelGGL(k_build_inputs, dim3(grid1d(NHW)), dim3(TPB), 0,
                     cur_stream(), raw.contiguous().data_ptr<float>(),
                     wb.contiguous().data_ptr<float>(),
                     ce.contiguous().data_ptr<float>(),
                     gc.contiguous().data_ptr<float>(),
                     (bf16_t*)cmg_in.data_ptr(), (bf16_t*)rwb.data_ptr(),
                     (bf16_t*)rce.data_ptr(), (bf16_t*)rgc.data_ptr(), NHW,
                     HW);
  HIP_CHECK_LAST();
  return {cmg_in, rwb, rce, rgc};
}

std::vector<at::Tensor> build_inputs_u8(const at::Tensor& raw,
                                        const at::Tensor& wb,
                                        const at::Tensor& ce,
                                        const at::Tensor& gc) {
  TORCH_CHECK(raw.is_cuda() && raw.dtype() == at::kByte && raw.dim() == 4 &&
              raw.size(3) == 3, "raw must be (N,H,W,3) uint8 CUDA");
  const long N = raw.size(0), H = raw.size(1), W = raw.size(2);
  const long NHW = N * H * W;
  auto opts = raw.options().dtype(at::kBFloat16);
  auto cmg_in = at::empty({N, H, W, 16}, opts);
  auto rwb = at::empty({N, H, W, 16}, opts);
  auto rce = at::empty({N, H, W, 16}, opts);
  auto rgc = at::empty({N, H, W, 16}, opts);
  hipLaunchKernelGGL(k_build_inputs_u8, dim3(grid1d(NHW)), dim3(TPB), 0,
                     cur_stream(), raw.contiguous().data_ptr<uint8_t>(),
                     wb.contiguous().data_ptr<uint8_t>(),
                     ce.contiguous().data_ptr<uint8_t>(),
                     gc.contiguous().data_ptr<uint8_t>(),
                     (bf16_t*)cmg_in.data_ptr(), (bf16_t*)rwb.data_ptr(),
                     (bf16_t*)rce.data_ptr(), (bf16_t*)rgc.data_ptr(), NHW);
  HIP_CHECK_LAST();
  return {cmg_in, rwb, rce, rgc};
}

at::Tensor u8_to_nhwc(const at::Tensor& x, int64_t Cp) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == at::kByte && x.dim() == 4 &&
              x.size(3) == 3);
  const long N = x.size(0), H = x.size(1), W = x.size(2);
  auto y = at::empty({N, H, W, Cp}, x.options().dtype(at::kBFloat16));
  const long NHW = N * H * W;
  hipLaunchKernelGGL(k_u8_to_nhwc, dim3(grid1d(NHW)), dim3(TPB), 0,
                     cur_stream(), x.contiguous().data_ptr<uint8_t>(),
                     (bf16_t*)y.data_ptr(), NHW, (int)Cp);
  HIP_CHECK_LAST();
  return y;
}

at::Tensor act_bwd_bias(const at::Tensor& dy, const at::Tensor& y,
                        int64_t act, at::Tensor& db) {
  TORCH_CHECK(dy.is_cuda() && dy.dtype() == at::kBFloat16 && dy.dim() == 4);
  TORCH_CHECK(db.is_cuda() && db.dtype() == at::kFloat);
  const int Kp = (int)dy.size(3);
  TORCH_CHECK(Kp <= 128 && (Kp & (Kp - 1)) == 0, "Kp power of two <= 128");
  const int K = (int)db.size(0);
  auto dpre = at::empty_like(dy);
  const long n = dy.numel();
  TORCH_CHECK(n % 8 == 0);
  const long n8 = n / 8;
  // blocks*blockDim*8 must be divisible by Kp: 256*8 = 2048 is divisible
  // by every Kp <= 128, so any block count works
  const int blocks = (int)std::min<long>(512, (n8 + 255) / 256);
  auto part = at::empty({blocks, (long)Kp},
                        dy.options().dtype(at::kFloat));
  hipLaunchKernelGGL(k_act_bwd_bias, dim3(blocks), dim3(TPB), 0,
                     cur_stream(), (const bf16_t*)dy.data_ptr(),
                     (const bf16_t*)y.data_ptr(), (bf16_t*)dpre.data_ptr(),
                     part.data_ptr<float>(), n8, Kp, (int)act);
  HIP_CHECK_LAST();
  hipLaunchKernelGGL(k_abb_reduce, dim3(K), dim3(256), 0, cur_stream(),
                     part.data_ptr<float>(), db.data_ptr<float>(), Kp, K,
                     blocks);
  HIP_CHECK_LAST();
  return dpre;
}

at::Tensor normalize_nhwc_fwd(const at::Tensor& x) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == at::kBFloat16 && x.dim() == 4);
  const long Cp = x.size(3);
  const long NHW = x.numel() / Cp;
  auto y = at::empty_like(x);
  hipLaunchKernelGGL(k_normalize_nhwc_fwd, dim3(grid1d(NHW)), dim3(TPB), 0,
                     cur_stream(), (const bf16_t*)x.contiguous().data_ptr(),
                     (bf16_t*)y.data_ptr(), NHW, (int)Cp);
  HIP_CHECK_LAST();
  return y;
}

at::Tensor normalize_nhwc_bwd(const at::Tensor& dy) {
  TORCH_CHECK(dy.is_cuda() && dy.dtype() == at::kBFloat16 && dy.dim() == 4);
  const long Cp = dy.size(3);
  const long NHW = dy.numel() / Cp;
  auto dx = at::empty_like(dy);
  hipLaunchKernelGGL(k_normalize_nhwc_bwd, dim3(grid1d(NHW)), dim3(TPB), 0,
                     cur_stream(), (const bf16_t*)dy.contiguous().data_ptr(),
                     (bf16_t*)dx.data_ptr(), NHW, (int)Cp);
  HIP_CHECK_LAST();
  return dx;
}

at::Tensor nchw_to_nhwc(const at::Tensor& x, int64_t Cp) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == at::kFloat && x.dim() == 4);
  const long N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  auto y = at::empty({N, H, W, Cp}, x.options().dtype(at::kBFloat16));
  const long NHW = N * H * W;
  hipLaunchKernelGGL(k_nchw2nhwc,
                     dim3((unsigned)((NHW + 255) / 256), 1, (unsigned)(Cp / 16)),
                     dim3(256), 0,
                     cur_stream(), x.contiguous().data_ptr<float>(),
                     (bf16_t*)y.data_ptr(), NHW, H * W, (int)C, (int)Cp);
  HIP_CHECK_LAST();
  return y;
}

at::Tensor nhwc_to_nchw(const at::Tensor& x, int64_t C) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == at::kBFloat16 && x.dim() == 4);
  const long N = x.size(0), H = x.size(1), W = x.size(2), Cp = x.size(3);
  auto y = at::empty({N, C, H, W}, x.options().dtype(at::kFloat));
  const long NHW = N * H * W;
  hipLaunchKernelGGL(k_nhwc2nchw,
                     dim3((unsigned)((NHW + 255) / 256), 1, (unsigned)(Cp / 16)),
                     dim3(256), 0,
                     cur_stream(), (const bf16_t*)x.contiguous().data_ptr(),
                     y.data_ptr<float>(), NHW, H * W, (int)C, (int)Cp);
  HIP_CHECK_LAST();
  return y;
}

at::Tensor fusion_fwd(const at::Tensor& maps, const at::Tensor& rwb,
                      const at::Tensor& rce, const at::Tensor& rgc) {
  const long N = maps.size(0), H = maps.size(1), W = maps.size(2);
  const long NHW = N * H * W;
  auto out = at::empty_like(maps);
  hipLaunchKernelGGL(k_fusion_fwd, dim3(grid1d(NHW)), dim3(TPB), 0,
                     cur_stream(), (const bf16_t*)maps.data_ptr(),
                     (const bf16_t*)rwb.data_ptr(),
                     (const bf16_t*)rce.data_ptr(),
                     (const bf16_t*)rgc.data_ptr(), (bf16_t*)out.data_ptr(),
                     NHW);
  HIP_CHECK_LAST();
  return out;
}

std::vector<at::Tensor> fusion_bwd(const at::Tensor& dout,
                                   const at::Tensor& maps,
                                   const at::Tensor& rwb,
                                   const at::Tensor& rce,
                                   const at::Tensor& rgc) {
  const long N = maps.size(0), H = maps.size(1), W = maps.size(2);
  const long NHW = N * H * W;
  auto dmaps = at::empty_like(maps);
  auto drwb = at::empty_like(maps);
  auto drce = at::empty_like(maps);
  auto drgc = at::empty_like(maps);
  hipLaunchKernelGGL(k_fusion_bwd, dim3(grid1d(NHW)), dim3(TPB), 0,
                     cur_stream(), (const bf16_t*)dout.data_ptr(),
                     (const bf16_t*)maps.data_ptr(),
                     (const bf16_t*)rwb.data_ptr(),
                     (const bf16_t*)rce.data_ptr(),
                     (const bf16_t*)rgc.data_ptr(), (bf16_t*)dmaps.data_ptr(),
                     (bf16_t*)drwb.data_ptr(), (bf16_t*)drce.data_ptr(),
                     (bf16_t*)drgc.data_ptr(), NHW);
  HIP_CHECK_LAST();
  return {dmaps, drwb, drce, drgc};
}

at::Tensor act_bwd(const at::Tensor& dy, const at::Tensor& y, int64_t act) {
  auto dpre = at::empty_like(dy);
  const long n = dy.numel();
  hipLaunchKernelGGL(k_act_bwd, dim3(grid1d((n + 7) / 8)), dim3(TPB), 0,
                     cur_stream(),
                     (const bf16_t*)dy.data_ptr(),
                     (const bf16_t*)y.data_ptr(), (bf16_t*)dpre.data_ptr(),
                     n, (int)act);
  HIP_CHECK_LAST();
  return dpre;
}

at::Tensor normalize_vgg_fwd(const at::Tensor& x, int64_t Cp) {
  const long N = x.size(0), H = x.size(2), W = x.size(3);
  const long NHW = N * H * W;
  auto y = at::empty({N, H, W, Cp}, x.options().dtype(at::kBFloat16));
  hipLaunchKernelGGL(k_normalize_vgg_fwd, dim3(grid1d(NHW)), dim3(TPB), 0,
                     cur_stream(), x.contiguous().data_ptr<float>(),
                     (bf16_t*)y.data_ptr(), NHW, H * W, (int)Cp);
  HIP_CHECK_LAST();
  return y;
}

at::Tensor normalize_vgg_bwd(const at::Tensor& dy, int64_t C) {
  TORCH_CHECK(C == 3);
  const long N = dy.size(0), H = dy.size(1), W = dy.size(2),
             Cp = dy.size(3);
  auto dx = at::empty({N, C, H, W}, dy.options().dtype(at::kFloat));
  const long NHW = N * H * W;
  hipLaunchKernelGGL(k_normalize_vgg_bwd, dim3(grid1d(NHW)), dim3(TPB), 0,
                     cur_stream(), (const bf16_t*)dy.contiguous().data_ptr(),
                     dx.data_ptr<float>(), NHW, H * W, (int)Cp);
  HIP_CHECK_LAST();
  return dx;
}

at::Tensor sqdiff255_sum(const at::Tensor& a, const at::Tensor& b,
                         int64_t Clog) {
  TORCH_CHECK(a.sizes() == b.sizes());
  auto out = at::zeros({}, a.options().dtype(at::kDouble));
  const long Cp = a.size(3);
  if (Clog == Cp) {
    const long n = a.numel();
    hipLaunchKernelGGL(k_sqdiff255_flat, dim3(grid1d(n)), dim3(TPB), 0,
                       cur_stream(), (const bf16_t*)a.data_ptr(),
                       (const bf16_t*)b.data_ptr(), out.data_ptr<double>(),
                       n);
  } else {
    const long NHW = a.numel() / Cp;
    hipLaunchKernelGGL(k_sqdiff255_pix, dim3(grid1d(NHW)), dim3(TPB), 0,
                       cur_stream(), (const bf16_t*)a.data_ptr(),
                       (const bf16_t*)b.data_ptr(), out.data_ptr<double>(),
                       NHW, (int)Clog, (int)Cp);
  }
  HIP_CHECK_LAST();
  return out;
}

at::Tensor sqdiff255_bwd(const at::Tensor& a, const at::Tensor& b,
                         const at::Tensor& gscale, double sign) {
  auto da = at::empty_like(a);
  const long n = a.numel();
  hipLaunchKernelGGL(k_sqdiff255_bwd, dim3(grid1d(n)), dim3(TPB), 0,
                     cur_stream(), (const bf16_t*)a.data_ptr(),
                     (const bf16_t*)b.data_ptr(), gscale.data_ptr<float>(),
                     (bf16_t*)da.data_ptr(), n, (float)sign);
  HIP_CHECK_LAST();
  return da;
}

at::Tensor out_to_u8(const at::Tensor& x) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == at::kBFloat16 && x.dim() == 4);
  const long N = x.size(0), H = x.size(1), W = x.size(2), Cp = x.size(3);
  auto y = at::empty({N, H, W, 3}, x.options().dtype(at::kByte));
  const long NHW = N * H * W;
  hipLaunchKernelGGL(k_out_to_u8, dim3(grid1d(NHW)), dim3(TPB), 0,
                     cur_stream(), (const bf16_t*)x.contiguous().data_ptr(),
                     y.data_ptr<uint8_t>(), NHW, (int)Cp);
  HIP_CHECK_LAST();
  return y;
}

at::Tensor u8_to_nchw(const at::Tensor& x) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == at::kByte && x.dim() == 4 &&
              x.size(3) == 3);
  const long N = x.size(0), H = x.size(1), W = x.size(2);
  auto y = at::empty({N, 3, H, W}, x.options().dtype(at::kFloat));
  const long NHW = N * H * W;
  hipLaunchKernelGGL(k_u8_to_nchw, dim3(grid1d(NHW)), dim3(TPB), 0,
                     cur_stream(), x.contiguous().data_ptr<uint8_t>(),
                     y.data_ptr<float>(), NHW, H * W);
  HIP_CHECK_LAST();
  return y;
}

void adam_step(at::Tensor& p, const at::Tensor& g, at::Tensor& m,
               at::Tensor& v, const at::Tensor& lr_buf, double b1, double b2,
               double eps, at::Tensor& step_buf) {
  TORCH_CHECK(p.is_cuda() && p.dtype() == at::kFloat);
  TORCH_CHECK(lr_buf.dtype() == at::kFloat && step_buf.dtype() == at::kInt);
  const long n = p.numel();
  hipLaunchKernelGGL(k_adam_tick, dim3(1), dim3(64), 0, cur_stream(),
                     step_buf.data_ptr<int>());
  hipLaunchKernelGGL(k_adam, dim3(grid1d(n)), dim3(TPB), 0, cur_stream(),
                     p.data_ptr<float>(), g.data_ptr<float>(),
                     m.data_ptr<float>(), v.data_ptr<float>(), n,
                     lr_buf.data_ptr<float>(), (float)b1, (float)b2,
                     (float)eps, step_buf.data_ptr<int>());
  HIP_CHECK_LAST();
}
