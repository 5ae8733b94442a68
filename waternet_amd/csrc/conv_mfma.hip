// MFMA implicit-GEMM convolution engine for gfx950 (CDNA4).
//
// Replaces the reference's implicit cuDNN conv launches (SURVEY §2.2
// K2-K9, K12-K14, K17) with hand-written NHWC bf16 kernels:
//   - k_conv_igemm<BN,KS,SPLITK>: stride-1 same-pad conv fwd as implicit
//     GEMM (M = N*H*W rows, N = output channels, K = R*S*Cp) on
//     v_mfma_f32_16x16x32_bf16 with fused bias + ReLU/Sigmoid epilogue and
//     global_load_lds staging; SPLITK slices the K loop over gridDim.z into
//     fp32 slabs (3-deep LDS ring, counted vmcnt) for small-M shapes.
//     Also runs dgrad (conv of dY with rotated/transposed packed weights).
//   - k_conv_igemm8<BN,KS>: 8-wave 256-row phase-split variant for big-M
//     shapes (setprio-wrapped MFMA phases, counted vmcnt).
//   - k_conv_wgrad<KS,BK,CH>: weight gradient as reduction GEMM over M,
//     m-major LDS images + ds_read_b64_tr_b16 hardware transpose reads,
//     fp32 atomic accumulation into the NCHW fp32 grad tensor.
//   - k_wgrad_smallk: VALU outer-product path for K <= 4 output convs.
//   - k_bias_grad(+_reduce): slab-partial column sums of dY.
//   - k_pack_fwd / k_pack_dgrad / k_pack_all: NCHW fp32 masters -> bf16
//     packed [Kp][R*S*Cp] (fwd) and rotated [Cp][R*S*Kp] (dgrad) layouts.
//
// Design notes (MI355X):
//   - igemm LDS tiles are FRAGMENT-MAJOR: slot (mf, kb, i) holds the 8
//     bf16 the MFMA lane (kb,i) consumes, so the lane-linear glds image and
//     ds_read_b128 fragment loads are bank-conflict-free without swizzles.
//   - Cp (physical channels) is a power of two >= 16; pad channels are
//     zero by construction everywhere, so no channel masking in the GEMM.
//
// WN_MFMA_KMAP: fragment K mapping, measured by probe_mfma.hip — 0 on this
// hardware (lane group g consumes reduction elements k = g*8+e).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

#ifndef WN_MFMA_KMAP
#define WN_MFMA_KMAP 0
#endif

// ---------------------------------------------------------------------------
// Forward / dgrad implicit GEMM
// ---------------------------------------------------------------------------

template <int BN, int KS, bool SPLITK = false>
__global__ __launch_bounds__(256, 2) void k_conv_igemm(
    const bf16_t* __restrict__ X,   // (N, H, W, Cp)
    const bf16_t* __restrict__ Wp,  // [Kp][RS*Cp] k-major packed
    const float* __restrict__ Bias, // [>=Klog] or nullptr
    bf16_t* __restrict__ Y,         // (N, H, W, Kp)
    int N, int H, int W, int Cp, int log2Cp, int Kp, int Klog, int act,
    const bf16_t* __restrict__ Zero16,    // 16 B of zeros (halo/pad source)
    float* __restrict__ Y32 = nullptr) {  // SPLITK: fp32 partial slabs.
                                        // NOT pre-zeroed (at::empty): every
                                        // slice block fully stores its m<M
                                        // rows, and finalize only reads
                                        // i < M*Kp — an epilogue edit that
                                        // skips stores must zero-fill first.
  constexpr int BM = 128;
  constexpr int PAD = KS / 2;
  constexpr int RS = KS * KS;
  constexpr int WM = (BN >= 64) ? 2 : 4;
  constexpr int WN = 4 / WM;
  constexpr int FM = BM / WM / 16;  // fragments per wave in M
  constexpr int FN = BN / WN / 16;  // fragments per wave in N
  constexpr int SLOTS_B = BN * 4;
  constexpr int NBS = (SLOTS_B + 255) / 256;  // B slots per thread (1 or 2)
  constexpr int LB = NBS * 256 * 8;  // B buffer elems (>= SLOTS_B*8, so the
                                     // glds of invalid slots lands in-pad)
  const int KG = RS * Cp;
  const long M = (long)N * H * W;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  // SPLITK uses a 5-deep buffer ring (see below); the 2-phase path uses
  // 2 stage buffers of DK 32-K steps each (DK=2 for the narrow tiles).
  constexpr int NBUF = SPLITK ? 5 : ((BN <= 32) ? 4 : 2);
  bf16_t* lA = reinterpret_cast<bf16_t*>(smem);              // NBUF x BM*32
  bf16_t* lB = lA + NBUF * BM * 32;                          // NBUF x LB

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid / WN;
  const int wc = wid % WN;
  const long m0 = (long)blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;

  // ---- per-thread A-slot geometry (2 slots) ----
  int aOy[2], aOx[2], aKb[2];
  long aRowBase[2];  // (n*H) term
  bool aMv[2];
  const int HW = H * W;
#pragma unroll
  for (int s = 0; s < 2; ++s) {
    int slot = tid + s * 256;
    int i = slot & 15, kb = (slot >> 4) & 3;
    int mf = slot >> 6;
    long m = m0 + mf * 16 + i;
    aMv[s] = m < M;
    long mm = aMv[s] ? m : 0;
    int n = (int)(mm / HW);
    int rem = (int)(mm - (long)n * HW);
    aOy[s] = rem / W;
    aOx[s] = rem - aOy[s] * W;
    aRowBase[s] = (long)n * H;
    aKb[s] = kb;
  }
  // ---- per-thread B-slot geometry ----
  int bK[NBS], bKb[NBS];
  bool bV[NBS];
#pragma unroll
  for (int s = 0; s < NBS; ++s) {
    int slot = tid + s * 256;
    bV[s] = slot < SLOTS_B;
    int j = slot & 15, kb = (slot >> 4) & 3;
    int nf = (slot >> 6);
    bK[s] = n0 + nf * 16 + j;
    bKb[s] = kb;
  }

  static_assert(WN_MFMA_KMAP == 0, "glds staging assumes KMAP 0");

  // Stage one K-step's A and B tiles straight into LDS buffer `buf` with
  // global_load_lds (16 B per lane, lane-linear LDS image — guide §5
  // "common mistake 1": width-16 glds is the staging lever). Out-of-range
  // lanes (conv halo, channel pad, M tail) read from Zero16 instead —
  // every lane always issues its DMA so the image is fully defined.
  auto stage = [&](int buf, int k0) {
#pragma unroll
    for (int s = 0; s < 2; ++s) {
      const bf16_t* src = Zero16;
      int rsc = k0 + aKb[s] * 8;
      if (aMv[s] && rsc < KG) {
        int tap = rsc >> log2Cp;
        int c = rsc & (Cp - 1);
        int dy = tap / KS, dx = tap - (tap / KS) * KS;
        int iy = aOy[s] + dy - PAD, ix = aOx[s] + dx - PAD;
        if (iy >= 0 && iy < H && ix >= 0 && ix < W)
          src = X + (((aRowBase[s] + iy) * W + ix) << log2Cp) + c;
      }
      __builtin_amdgcn_global_load_lds(
          src, lA + buf * (BM * 32) + (tid + s * 256) * 8, 16, 0, 0);
    }
#pragma unroll
    for (int s = 0; s < NBS; ++s) {
      const bf16_t* src = Zero16;
      int rsc = k0 + bKb[s] * 8;
      if (bV[s] && bK[s] < Kp && rsc < KG)
        src = Wp + (long)bK[s] * KG + rsc;
      __builtin_amdgcn_global_load_lds(
          src, lB + buf * LB + (tid + s * 256) * 8, 16, 0, 0);
    }
  };

  f32x4 acc[FM][FN];
#pragma unroll
  for (int a = 0; a < FM; ++a)
#pragma unroll
    for (int b = 0; b < FN; ++b) acc[a][b] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int nkAll = (KG + 31) / 32;
  // SPLITK: z-slice [ks0, nk) of the K loop; else the whole range. Floor
  // slicing keeps every slice non-empty for gz <= nkAll, so every fp32
  // slab is fully written (the finalize pass sums all of them).
  int ks0 = 0, nk = nkAll;
  if (SPLITK) {
    ks0 = (int)((long)blockIdx.z * nkAll / gridDim.z);
    nk = (int)((long)(blockIdx.z + 1) * nkAll / gridDim.z);
  }
  const int lg = lane >> 4;   // fragment k-group
  const int li = lane & 15;   // fragment row/col

  if constexpr (SPLITK) {
    // Split-K shapes run at ~1 block/CU (small M, chip filled by K
    // slices): the per-step vmcnt(0) drain is fully exposed there, so use
    // a 3-buffer glds pipeline with COUNTED vmcnt + raw barriers (guide
    // §5 "Pipelining across barriers": +40-83% in the 1-block/CU regime;
    // measured NEGATIVE at the big-M shapes' 5-blocks/CU occupancy, which
    // keep the simpler 2-phase below).
    constexpr int GPS = 2 + NBS;  // glds per stage per thread (3 or 4)
    // wait until at most `tiles`' worth of glds remain outstanding
    auto wait_tiles = [&](int tiles) {
      if constexpr (GPS == 3) {
        switch (tiles) {
          case 3: asm volatile("s_waitcnt vmcnt(9)" ::: "memory"); break;
          case 2: asm volatile("s_waitcnt vmcnt(6)" ::: "memory"); break;
          case 1: asm volatile("s_waitcnt vmcnt(3)" ::: "memory"); break;
          default: asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        }
      } else {
        switch (tiles) {
          case 3: asm volatile("s_waitcnt vmcnt(12)" ::: "memory"); break;
          case 2: asm volatile("s_waitcnt vmcnt(8)" ::: "memory"); break;
          case 1: asm volatile("s_waitcnt vmcnt(4)" ::: "memory"); break;
          default: asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        }
      }
    };
    // prologue: up to 4 tiles in flight, wait for tile ks0 only
    for (int t = 0; t < 4 && ks0 + t < nk; ++t) stage(t, (ks0 + t) * 32);
    wait_tiles(min(nk - ks0 - 1, 3));
    __builtin_amdgcn_s_barrier();
    for (int ks = ks0; ks < nk; ++ks) {
      const int b = (ks - ks0) % 5;
      if (ks + 4 < nk) stage((b + 4) % 5, (ks + 4) * 32);
      bf16x8 aF[FM], bF[FN];
#pragma unroll
      for (int fm = 0; fm < FM; ++fm)
        aF[fm] = *reinterpret_cast<const bf16x8*>(
            lA + b * (BM * 32) + (((wr * FM + fm) * 4 + lg) * 16 + li) * 8);
#pragma unroll
      for (int fn = 0; fn < FN; ++fn)
        bF[fn] = *reinterpret_cast<const bf16x8*>(
            lB + b * LB + (((wc * FN + fn) * 4 + lg) * 16 + li) * 8);
#pragma unroll
      for (int fm = 0; fm < FM; ++fm)
#pragma unroll
        for (int fn = 0; fn < FN; ++fn)
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              aF[fm], bF[fn], acc[fm][fn], 0, 0, 0);
      if (ks + 1 < nk) {
        wait_tiles(min(nk - ks - 2, 3));  // tile ks+1 landed
        __builtin_amdgcn_s_barrier();
      }
    }
  } else {
    // 2-phase glds pipeline (guide §5 "minimum 2-phase"): issue next
    // tile's DMA first, ds_read + MFMA the current buffer, one
    // vmcnt(0)+barrier per staged tile (the __syncthreads drains the DMA).
    // Narrow tiles (BN<=32) do only FM*FN<=4 MFMAs per 32-K step, so they
    // stage DK=2 steps per barrier to amortize it (K tails are zero-filled
    // by the staging, so no edge code).
    constexpr int DK = (BN <= 32) ? 2 : 1;
#pragma unroll
    for (int h = 0; h < DK; ++h) stage(h, (ks0 + h) * 32);
    __syncthreads();
    int cur = 0;
    const int nk2 = (nk - ks0 + DK - 1) / DK;
    for (int ks2 = 0; ks2 < nk2; ++ks2) {
      if (ks2 + 1 < nk2) {
#pragma unroll
        for (int h = 0; h < DK; ++h)
          stage((cur ^ 1) * DK + h, (ks0 + (ks2 + 1) * DK + h) * 32);
      }
#pragma unroll
      for (int h = 0; h < DK; ++h) {
        bf16x8 aF[FM], bF[FN];
#pragma unroll
        for (int fm = 0; fm < FM; ++fm) {
          int mfG = wr * FM + fm;
          aF[fm] = *reinterpret_cast<const bf16x8*>(
              lA + (cur * DK + h) * (BM * 32) +
              ((mfG * 4 + lg) * 16 + li) * 8);
        }
#pragma unroll
        for (int fn = 0; fn < FN; ++fn) {
          int nfG = wc * FN + fn;
          bF[fn] = *reinterpret_cast<const bf16x8*>(
              lB + (cur * DK + h) * LB + ((nfG * 4 + lg) * 16 + li) * 8);
        }
#pragma unroll
        for (int fm = 0; fm < FM; ++fm)
#pragma unroll
          for (int fn = 0; fn < FN; ++fn)
            acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                aF[fm], bF[fn], acc[fm][fn], 0, 0, 0);
      }
      if (ks2 + 1 < nk2) __syncthreads();
      cur ^= 1;
    }
  }

  // ---- epilogue: bias + activation (+ pad-channel zeroing), or fp32
  //      partial-sum atomics when this launch is a split-K slice ----
  const int lr4 = (lane >> 4) * 4;
#pragma unroll
  for (int fn = 0; fn < FN; ++fn) {
    const int k = n0 + (wc * FN + fn) * 16 + li;
    if (k >= Kp) continue;
    const float bv = (Bias != nullptr && k < Klog) ? Bias[k] : 0.f;
    const bool kpad = k >= Klog;
#pragma unroll
    for (int fm = 0; fm < FM; ++fm) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long m = m0 + (wr * FM + fm) * 16 + lr4 + r;
        if (m >= M) continue;
        if (SPLITK) {
          // per-slice fp32 slab, plain stores (no atomics — the finalize
          // pass reduces over gridDim.z slabs)
          Y32[((long)blockIdx.z * M + m) * Kp + k] = acc[fm][fn][r];
        } else {
          float v = acc[fm][fn][r] + bv;
          if (act == ACT_RELU) v = fmaxf(v, 0.f);
          else if (act == ACT_SIGMOID) v = 1.f / (1.f + __expf(-v));
          if (kpad) v = 0.f;
          Y[m * Kp + k] = f2bf(v);
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// 8-wave 256xBN phase-split conv igemm for BIG-M shapes (guide §5 T3+T5:
// wave role diversity + setprio + counted vmcnt; the 4-wave 2-phase kernel
// above is schedule-bound at ~480 TF on these shapes — PMC r06/r18: ~58%
// WAIT_ANY with MFMA, TA and HBM all unsaturated).
//   - 512 threads = 8 waves as 4(M)x2(N); per-wave output 64 x BN/2.
//   - 3-deep LDS ring, glds staging (GPS=3 per thread per tile), counted
//     s_waitcnt vmcnt leaves the next tile's DMA in flight across barriers.
//   - two MFMA phases per K-step with s_setprio(1) around the matrix math.
// ---------------------------------------------------------------------------

// VAR: 0 = 2 setprio phases / 3-ring (default); 1 = no setprio;
// 2 = 4 phases; 3 = 4-deep ring (2 tiles in flight). Runtime-selectable
// via WN_IGEMM8_VAR for within-box A/B (guide §5.4 rule 24).
// M32: v_mfma_f32_32x32x16_bf16 fragments (half the MFMA instruction
// count at the higher 32x32 pipe rate; +5.5% proven on the wgrad).
template <int BN, int KS, int VAR = 0, bool M32 = false>
__global__ __launch_bounds__(512, 1) void k_conv_igemm8(
    const bf16_t* __restrict__ X,   // (N, H, W, Cp)
    const bf16_t* __restrict__ Wp,  // [Kp][RS*Cp]
    const float* __restrict__ Bias,
    bf16_t* __restrict__ Y,         // (N, H, W, Kp)
    int N, int H, int W, int Cp, int log2Cp, int Kp, int Klog, int act,
    const bf16_t* __restrict__ Zero16) {
  static_assert(WN_MFMA_KMAP == 0, "glds staging assumes KMAP 0");
  constexpr int BM = 256;
  constexpr int PAD = KS / 2;
  constexpr int RS = KS * KS;
  constexpr int WMW = 4, WNW = 2;       // wave grid
  constexpr int FM = BM / WMW / 16;     // 4 fragments in M per wave
  constexpr int FN = BN / WNW / 16;     // 4 (BN=128) / 2 (BN=64)
  constexpr int ASLOT = BM * 32 / 8 / 512;   // = 2
  constexpr int SLOTS_B = BN * 4;
  constexpr int BSLOT = (SLOTS_B + 511) / 512;  // = 1
  constexpr int LA = BM * 32;           // A buffer elems
  constexpr int LB8 = 512 * 8;          // B buffer elems (>= SLOTS_B*8)
  const int KG = RS * Cp;
  const long M = (long)N * H * W;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16_t* lA = reinterpret_cast<bf16_t*>(smem);   // 3 x LA
  bf16_t* lB = lA + 3 * LA;                       // 3 x LB8

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 1;
  const int wc = wid & 1;
  const long m0 = (long)blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;
  const int HW = H * W;

  // A slots: the lane-linear LDS image IS the fragment layout, so the
  // slot->(m row, k offset) decode differs per MFMA shape:
  //   16x16x32: slot = (mf16, kb, i16) -> m = mf16*16+i16, kOff = kb*8
  //   32x32x16: slot = (mf32, t, g, i32) -> m = mf32*32+i32,
  //             kOff = t*16 + g*8  (lane l reads k = (l>>5)*8+e)
  int aOy[ASLOT], aOx[ASLOT], aKb[ASLOT];
  long aRowBase[ASLOT];
  bool aMv[ASLOT];
#pragma unroll
  for (int s = 0; s < ASLOT; ++s) {
    int slot = tid + s * 512;
    int mrow, koff;
    if constexpr (M32) {
      mrow = (slot >> 7) * 32 + (slot & 31);
      koff = ((slot >> 6) & 1) * 16 + ((slot >> 5) & 1) * 8;
    } else {
      mrow = (slot >> 6) * 16 + (slot & 15);
      koff = ((slot >> 4) & 3) * 8;
    }
    long m = m0 + mrow;
    aMv[s] = m < M;
    long mm = aMv[s] ? m : 0;
    int n = (int)(mm / HW);
    int rem = (int)(mm - (long)n * HW);
    aOy[s] = rem / W;
    aOx[s] = rem - aOy[s] * W;
    aRowBase[s] = (long)n * H;
    aKb[s] = koff;
  }
  int bK[BSLOT], bKb[BSLOT];
  bool bV[BSLOT];
#pragma unroll
  for (int s = 0; s < BSLOT; ++s) {
    int slot = tid + s * 512;
    bV[s] = slot < SLOTS_B;
    int col, koff;
    if constexpr (M32) {
      col = (slot >> 7) * 32 + (slot & 31);
      koff = ((slot >> 6) & 1) * 16 + ((slot >> 5) & 1) * 8;
    } else {
      col = (slot >> 6) * 16 + (slot & 15);
      koff = ((slot >> 4) & 3) * 8;
    }
    bK[s] = n0 + col;
    bKb[s] = koff;
  }

  auto stage = [&](int buf, int ks) {
    const int k0 = ks * 32;
#pragma unroll
    for (int s = 0; s < ASLOT; ++s) {
      const bf16_t* src = Zero16;
      int rsc = k0 + aKb[s];
      if (aMv[s] && rsc < KG) {
        int tap = rsc >> log2Cp;
        int c = rsc & (Cp - 1);
        int dy = tap / KS, dx = tap - (tap / KS) * KS;
        int iy = aOy[s] + dy - PAD, ix = aOx[s] + dx - PAD;
        if (iy >= 0 && iy < H && ix >= 0 && ix < W)
          src = X + (((aRowBase[s] + iy) * W + ix) << log2Cp) + c;
      }
      __builtin_amdgcn_global_load_lds(
          src, lA + buf * LA + (tid + s * 512) * 8, 16, 0, 0);
    }
#pragma unroll
    for (int s = 0; s < BSLOT; ++s) {
      const bf16_t* src = Zero16;
      int rsc = k0 + bKb[s];
      if (bV[s] && bK[s] < Kp && rsc < KG)
        src = Wp + (long)bK[s] * KG + rsc;
      __builtin_amdgcn_global_load_lds(
          src, lB + buf * LB8 + (tid + s * 512) * 8, 16, 0, 0);
    }
  };

  constexpr int FM32 = BM / WMW / 32;       // 2
  constexpr int FN32 = (BN / WNW) / 32;     // 2 (BN=128) / 1 (BN=64)
  f32x4 acc[M32 ? 1 : FM][M32 ? 1 : FN];
  f32x16 acc32[M32 ? FM32 : 1][M32 ? FN32 : 1];
#pragma unroll
  for (int a = 0; a < (M32 ? FM32 : FM); ++a)
#pragma unroll
    for (int b = 0; b < (M32 ? FN32 : FN); ++b) {
      if constexpr (M32) {
#pragma unroll
        for (int e = 0; e < 16; ++e) acc32[a][b][e] = 0.f;
      } else {
        acc[a][b] = f32x4{0.f, 0.f, 0.f, 0.f};
      }
    }

  const int nk = (KG + 31) / 32;
  const int lg = lane >> 4, li = lane & 15;
  constexpr int GPS = ASLOT + BSLOT;  // 3 glds per thread per stage
  constexpr int RING = (VAR == 3) ? 4 : 3;
  constexpr int AHEAD = RING - 2;  // tiles left in flight at the wait
  constexpr bool PRIO = (VAR != 1);
  constexpr int NPH = (VAR == 2) ? 4 : 2;  // M-fragment phases per K-step

  auto wait_tiles = [&](int inflight) {
    if (inflight >= 2 && AHEAD == 2)
      asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    else if (inflight >= 1)
      asm volatile("s_waitcnt vmcnt(3)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  };
  static_assert(GPS == 3, "vmcnt immediates assume 3 glds per stage");

  for (int t = 0; t < RING - 1 && t < nk; ++t) stage(t, t);
  wait_tiles(min(nk - 1, RING - 2));
  __builtin_amdgcn_s_barrier();

  for (int ks = 0; ks < nk; ++ks) {
    const int b = ks % RING;
    if (ks + RING - 1 < nk) stage((b + RING - 1) % RING, ks + RING - 1);
    if constexpr (M32) {
      // lane-linear fragment reads: block (mf32*2+t)*2+g at lane*16B
      bf16x8 aF[FM32][2], bF[FN32][2];
#pragma unroll
      for (int t = 0; t < 2; ++t)
#pragma unroll
        for (int fn = 0; fn < FN32; ++fn)
          bF[fn][t] = *reinterpret_cast<const bf16x8*>(
              lB + b * LB8 + ((wc * FN32 + fn) * 2 + t) * 512 + lane * 8);
#pragma unroll
      for (int ph = 0; ph < 2; ++ph) {  // one M fragment per phase
#pragma unroll
        for (int t = 0; t < 2; ++t)
          aF[ph][t] = *reinterpret_cast<const bf16x8*>(
              lA + b * LA + ((wr * FM32 + ph) * 2 + t) * 512 + lane * 8);
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_sched_barrier(0);
        if constexpr (PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int t = 0; t < 2; ++t)
#pragma unroll
          for (int fn = 0; fn < FN32; ++fn)
            acc32[ph][fn] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                aF[ph][t], bF[fn][t], acc32[ph][fn], 0, 0, 0);
        if constexpr (PRIO) __builtin_amdgcn_s_setprio(0);
      }
    } else {
      bf16x8 aF[FM], bF[FN];
#pragma unroll
      for (int fn = 0; fn < FN; ++fn)
        bF[fn] = *reinterpret_cast<const bf16x8*>(
            lB + b * LB8 + (((wc * FN + fn) * 4 + lg) * 16 + li) * 8);
#pragma unroll
      for (int ph = 0; ph < NPH; ++ph) {
        constexpr int FPP = M32 ? 1 : FM / NPH;  // fragments per phase
#pragma unroll
        for (int fm = ph * FPP; fm < (ph + 1) * FPP; ++fm)
          aF[fm] = *reinterpret_cast<const bf16x8*>(
              lA + b * LA + (((wr * FM + fm) * 4 + lg) * 16 + li) * 8);
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_sched_barrier(0);
        if constexpr (PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int fm = ph * FPP; fm < (ph + 1) * FPP; ++fm)
#pragma unroll
          for (int fn = 0; fn < FN; ++fn)
            acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                aF[fm], bF[fn], acc[fm][fn], 0, 0, 0);
        if constexpr (PRIO) __builtin_amdgcn_s_setprio(0);
      }
    }
    if (ks + 1 < nk) {
      wait_tiles(min(nk - ks - 2, AHEAD));
      __builtin_amdgcn_s_barrier();
    }
  }

  if constexpr (M32) {
    // 32x32 D map: col = lane&31 (k-out), row = (r&3)+8*(r>>2)+4*(lane>>5)
#pragma unroll
    for (int fn = 0; fn < FN32; ++fn) {
      const int k = n0 + (wc * FN32 + fn) * 32 + (lane & 31);
      if (k >= Kp) continue;
      const float bv = (Bias != nullptr && k < Klog) ? Bias[k] : 0.f;
      const bool kpad = k >= Klog;
#pragma unroll
      for (int fm = 0; fm < FM32; ++fm) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
          long m = m0 + (wr * FM32 + fm) * 32 + row;
          if (m >= M) continue;
          float v = acc32[fm][fn][r] + bv;
          if (act == ACT_RELU) v = fmaxf(v, 0.f);
          else if (act == ACT_SIGMOID) v = 1.f / (1.f + __expf(-v));
          if (kpad) v = 0.f;
          Y[m * Kp + k] = f2bf(v);
        }
      }
    }
    return;
  }
  // epilogue: bias + activation + pad zeroing
  const int lr4 = lg * 4;
#pragma unroll
  for (int fn = 0; fn < FN; ++fn) {
    const int k = n0 + (wc * FN + fn) * 16 + li;
    if (k >= Kp) continue;
    const float bv = (Bias != nullptr && k < Klog) ? Bias[k] : 0.f;
    const bool kpad = k >= Klog;
#pragma unroll
    for (int fm = 0; fm < FM; ++fm) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long m = m0 + (wr * FM + fm) * 16 + lr4 + r;
        if (m >= M) continue;
        float v = acc[fm][fn][r] + bv;
        if (act == ACT_RELU) v = fmaxf(v, 0.f);
        else if (act == ACT_SIGMOID) v = 1.f / (1.f + __expf(-v));
        if (kpad) v = 0.f;
        Y[m * Kp + k] = f2bf(v);
      }
    }
  }
}

// Split-K finalize: Y = act(sum over gz slabs of Y32 + bias), pad zeroed.
// GZ is a template constant so all slab loads issue in parallel (the
// runtime-bounded loop serialized up to 8 dependent ~500-cycle loads per
// element — measured 9.5 us per call, ~1.5 us unrolled).
template <int GZ>
__global__ void k_splitk_finalize(const float* __restrict__ Y32,
                                  const float* __restrict__ Bias,
                                  bf16_t* __restrict__ Y, long M, int Kp,
                                  int Klog, int act) {
  const long total = M * Kp;
  for (long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
       i0 < total; i0 += (long)gridDim.x * blockDim.x * 4) {
    f32x4 vz[GZ];
#pragma unroll
    for (int z = 0; z < GZ; ++z)
      vz[z] = *reinterpret_cast<const f32x4*>(Y32 + z * total + i0);
    f32x4 v = vz[0];
#pragma unroll
    for (int z = 1; z < GZ; ++z)
#pragma unroll
      for (int e = 0; e < 4; ++e) v[e] += vz[z][e];
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      const long i = i0 + e;
      const int k = (int)(i % Kp);
      float w = v[e] + ((Bias != nullptr && k < Klog) ? Bias[k] : 0.f);
      if (act == ACT_RELU) w = fmaxf(w, 0.f);
      else if (act == ACT_SIGMOID) w = 1.f / (1.f + __expf(-w));
      if (k >= Klog) w = 0.f;
      Y[i] = f2bf(w);
    }
  }
}

// ---------------------------------------------------------------------------
// Weight gradient: dW[k][rsc] = sum_m dY[m][k] * X[m][rsc]
//
// v3: both GEMM operands are [reduction=m]-major in global memory (dY is
// (m,k), X gathers are (m,c)), i.e. TRANSPOSED relative to the MFMA
// fragment layout (lane needs 8 m for a fixed column). v2 transposed via
// 8x ds_write_b16 scatter, which lands 16-32 lanes on the same mod-32
// write bank (ds_write banks are (a/4)%32) — LDS-write-bound. v3 stores
// the tiles m-major with vector ds_write_b128 (conflict-free by kblk
// stride choice) and uses gfx950's ds_read_b64_tr_b16 hardware transpose
// read to deliver MFMA fragments (guide T10, the attention-V recipe).
//
// LDS image per operand: [col/16 kblk][m/4 mblk][4m][16col] subtiles,
// row-major inside a subtile; mblk stride 96 elems (2*96 dwords = 32 mod
// 64 so tr-read lane groups land on disjoint read banks), kblk stride
// 1552 elems (776 dwords = 8 mod 32 so the staging b128 writes of one
// 8-lane service group hit 8 distinct bank quads).
//
//   - BK x 128 output tile per block (BK = min(Kp,128)), 2x2 / 1x4 waves
//   - m-chunks of 64; DOUBLE-BUFFERED global_load_lds staging: the linear
//     chunk index of the tr image maps identically to tr_addr, so the DMA
//     writes the image lane-linearly while per-lane sources gather the
//     right 16 B of dY / halo-masked X (pad lanes read a zero buffer).
//     The next chunk's DMA issues before this chunk's MFMAs and drains at
//     the single trailing barrier.
//   - default MFMA is v_mfma_f32_32x32x16_bf16 (M32) with a per-kstep
//     counted-lgkmcnt tr-read interleave; WN_WGRAD_M32=0 selects the
//     16x16x32 path.
//   - split-m z dimension sized to fill the CUs; gy (rsc tiles) varies
//     fastest so co-resident blocks share dY/X chunks through L2.
//   - fp32 atomicAdd epilogue into the NCHW fp32 grad tensor (pre-zeroed).
// Requires the KMAP-0 fragment layout (lane group g consumes k = g*8+e).
// ---------------------------------------------------------------------------

constexpr int TR_MBS = 72;    // mblk stride (36 dwords; 2 mblks = 8 mod 64
                              // banks -> at most 2-way tr-read overlap, and
                              // the image fits 4 blocks/CU at CH=64)
constexpr int TR_KBS = 16 * TR_MBS + 16;  // kblk stride (584 dw = 8 mod 32:
                                          // conflict-free staging writes)

// Magic-multiply unsigned division by a launch-constant divisor d:
// q = (n * mul) >> 42 with mul = floor(2^42/d)+1 — exact for n*d < 2^42
// (holds for any image geometry we launch). Replaces the 64-bit v_div
// chains the m->(n,oy,ox) decode otherwise costs per chunk (PMC r06:
// wgrad SQ_WAIT_INST_ANY 47-75% with zero bank conflicts = serial VALU
// division RAW stalls).
struct MagicDiv {
  unsigned long long mul;
  static MagicDiv make(unsigned d) {
    return {(0x40000000000ULL / d) + 1};
  }
};
WN_DEVFN unsigned magic_div(unsigned n, unsigned long long mul) {
  return (unsigned)(((unsigned long long)n * mul) >> 42);
}

// elem offset of (m, col) inside a tr image
template <int KBS = TR_KBS>
WN_DEVFN int tr_addr(int m, int col) {
  return (col >> 4) * KBS + (m >> 2) * TR_MBS + (m & 3) * 16 + (col & 15);
}

typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;

// ds_read_b64_tr_b16: lane (l&15) of each 16-lane group receives column
// (l&15), rows 0..3, of the [4][16] row-major bf16 subtile whose 16
// 8-byte pieces the group's lanes address in canonical order (lane i ->
// subtile_base + i*8B). addr is an LDS byte address.
WN_DEVFN bf16x4 ds_tr16(unsigned addr) {
  bf16x4 v;
  // "memory" keeps the read below the staging stores / barrier (without it
  // hipcc hoists the first tr read above the LDS fill — measured garbage).
  asm volatile("ds_read_b64_tr_b16 %0, %1" : "=v"(v) : "v"(addr) : "memory");
  return v;
}

template <int KS, int BK, int CH = 64, bool M32 = false>
__global__ __launch_bounds__(256, 2) void k_conv_wgrad(
    const bf16_t* __restrict__ dY,  // (N,H,W,Kp)
    const bf16_t* __restrict__ X,   // (N,H,W,Cp)
    float* __restrict__ dW,         // (K, C, KS, KS) fp32, pre-zeroed
    int N, int H, int W, int Cp, int log2Cp, int Kp, int K, int C,
    int splitm, unsigned long long mulHW, unsigned long long mulW,
    const bf16_t* __restrict__ Zero16w) {
  static_assert(WN_MFMA_KMAP == 0, "wgrad staging assumes KMAP 0");
  constexpr int PAD = KS / 2;
  constexpr int RS = KS * KS;
  constexpr int BR = 128;           // rsc tile
  // per-chunk tr image kblk stride: CH/4 mblks x 96 elems + 16 pad, which
  // keeps the stride 8 mod 32 dwords (conflict-free staging writes)
  constexpr int KBS = CH / 4 * TR_MBS + 16;  // = TR_KBS at CH=64
  constexpr int WR = (BK >= 64) ? 2 : 1;   // wave rows (k dim)
  constexpr int WC = 4 / WR;               // wave cols (rsc dim)
  constexpr int FK = BK / WR / 16;  // k fragments per wave
  constexpr int FR = BR / WC / 16;  // rsc fragments per wave
  const int KG = RS * Cp;
  const long M = (long)N * H * W;
  const int HW = H * W;

  // glds staging: the tr image's linear 8-element chunk index sigma maps
  // IDENTICALLY to tr_addr (sigma*8 == tr_addr(m, col) for m = mblk*4 +
  // (r>>1), col = kblk*16 + (r&1)*8, sigma = kblk*CPK + mblk*9 + r), so
  // global_load_lds writes the image lane-linearly while each lane's
  // SOURCE gathers the right 16 B of dY/X (pad/halo lanes read Zero16).
  // Double-buffered: the next chunk's DMA issues before this chunk's
  // MFMAs and drains at the single trailing barrier.
  constexpr int CPK = KBS / 8;               // chunks per kblk
  constexpr int AC = (BK / 16) * CPK;        // A image chunks
  constexpr int BC = (BR / 16) * CPK;        // B image chunks
  constexpr int SA = (AC + 255) / 256;
  constexpr int SB = (BC + 255) / 256;
  constexpr int ABUF = SA * 256 * 8;         // elems per buffer (padded so
  constexpr int BBUF = SB * 256 * 8;         //  out-of-image slots land in-pad)

  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16_t* lA = reinterpret_cast<bf16_t*>(smem);   // 2 x ABUF
  bf16_t* lB = lA + 2 * ABUF;                     // 2 x BBUF

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid / WC;
  const int wc = wid % WC;
  const int kt0 = blockIdx.x * BK;
  const int rt0 = blockIdx.y * BR;

  // ---- static slot geometry (sigma -> image position -> source) ----
  int aM_[SA], aCol[SA];
  bool aPad[SA];
#pragma unroll
  for (int s = 0; s < SA; ++s) {
    const int sg = tid + s * 256;
    const int kblk = sg / CPK, c = sg % CPK;
    const int mblk = c / 9, r = c % 9;
    aPad[s] = (sg >= AC) || (r == 8) || (c >= (CH / 4) * 9);
    aM_[s] = mblk * 4 + (r >> 1);
    aCol[s] = kt0 + kblk * 16 + (r & 1) * 8;
  }
  int bM_[SB], bC_[SB], bDy[SB], bDx[SB];
  bool bPad[SB];
#pragma unroll
  for (int s = 0; s < SB; ++s) {
    const int sg = tid + s * 256;
    const int kblk = sg / CPK, c = sg % CPK;
    const int mblk = c / 9, r = c % 9;
    const int rsc = rt0 + kblk * 16 + (r & 1) * 8;
    bPad[s] = (sg >= BC) || (r == 8) || (c >= (CH / 4) * 9) || (rsc >= KG);
    bM_[s] = mblk * 4 + (r >> 1);
    const int tap = rsc >> log2Cp;
    bC_[s] = rsc & (Cp - 1);
    bDy[s] = tap / KS;
    bDx[s] = tap - bDy[s] * KS;
  }

  auto stage = [&](int buf, long mbase) {
#pragma unroll
    for (int s = 0; s < SA; ++s) {
      const bf16_t* src = Zero16w;
      const long m = mbase + aM_[s];
      if (!aPad[s] && m < M) src = dY + m * Kp + aCol[s];
      __builtin_amdgcn_global_load_lds(
          src, lA + buf * ABUF + (tid + s * 256) * 8, 16, 0, 0);
    }
#pragma unroll
    for (int s = 0; s < SB; ++s) {
      const bf16_t* src = Zero16w;
      const long m = mbase + bM_[s];
      if (!bPad[s] && m < M) {
        unsigned n = magic_div((unsigned)m, mulHW);
        unsigned rem = (unsigned)m - n * (unsigned)HW;
        unsigned oy = magic_div(rem, mulW);
        int ox = (int)(rem - oy * (unsigned)W);
        int iy = (int)oy + bDy[s] - PAD, ix = ox + bDx[s] - PAD;
        if (iy >= 0 && iy < H && ix >= 0 && ix < W)
          src = X + (((long)((int)n * H + iy) * W + ix) << log2Cp) + bC_[s];
      }
      __builtin_amdgcn_global_load_lds(
          src, lB + buf * BBUF + (tid + s * 256) * 8, 16, 0, 0);
    }
  };

  // M32 variant: v_mfma_f32_32x32x16_bf16 — half the MFMA instruction
  // count for the same MACs at the 2382-vs-2075 TF pipe rate (PMC shows
  // the 16x16 wgrad is MFMA-pipe-saturated). Wave tile stays 64x64
  // (BK>=64): FK32 x FR32 accumulators of 16 f32.
  constexpr int FK32 = (BK >= 64) ? BK / WR / 32 : 1;
  constexpr int FR32 = BR / WC / 32;
  f32x4 acc[M32 ? 1 : FK][M32 ? 1 : FR];
  f32x16 acc32[M32 ? FK32 : 1][M32 ? FR32 : 1];
#pragma unroll
  for (int a = 0; a < (M32 ? FK32 : FK); ++a)
#pragma unroll
    for (int b = 0; b < (M32 ? FR32 : FR); ++b) {
      if constexpr (M32) {
#pragma unroll
        for (int e = 0; e < 16; ++e) acc32[a][b][e] = 0.f;
      } else {
        acc[a][b] = f32x4{0.f, 0.f, 0.f, 0.f};
      }
    }

  const long nChunks = (M + CH - 1) / CH;
  const int lg = lane >> 4, li = lane & 15;
  // per-lane tr-read byte addresses for kstep 0 / mblk (2*lg); the second
  // 4-m half (e=4..7) is the next mblk at +2*TR_MBS bytes.
  const unsigned aTr0 = (unsigned)(unsigned long long)lA +
                        (unsigned)(2 * lg * TR_MBS + li * 4) * 2;
  const unsigned bTr0 = (unsigned)(unsigned long long)lB +
                        (unsigned)(2 * lg * TR_MBS + li * 4) * 2;
  // M32 per-lane bases: lane l reads column (l&31) => col-block (l>>4)&1,
  // reduction m = (l>>5)*8 + e => mblk offset (l>>5)*2 (+h)
  const unsigned trM32 =
      (unsigned)(((lane >> 4) & 1) * KBS + (lane >> 5) * 2 * TR_MBS +
                 (lane & 15) * 4) * 2;
  const unsigned aTrM = (unsigned)(unsigned long long)lA + trM32;
  const unsigned bTrM = (unsigned)(unsigned long long)lB + trM32;

  long chunk = blockIdx.z;
  int buf = 0;
  if (chunk < nChunks) {
    stage(0, chunk * CH);
    __syncthreads();  // drains the DMA (vmcnt 0) + barrier
  }
  for (; chunk < nChunks; chunk += splitm) {
    const long next = chunk + splitm;
    if (next < nChunks) stage(buf ^ 1, next * CH);  // DMA under the math
    const unsigned aOff = (unsigned)(buf * ABUF) * 2;
    const unsigned bOff = (unsigned)(buf * BBUF) * 2;
    if constexpr (M32) {
      // fine-grained per-kstep interleave: tr-read kstep k+1 while the
      // MFMAs of kstep k issue, with COUNTED lgkmcnt so waves drift apart
      // instead of entering barrier-aligned MFMA bursts together (PMC:
      // 61% SQ_WAIT_INST_ANY at only ~24% MFMA-pipe duty).
      constexpr int KST = CH / 16;  // 32x32x16 ksteps per chunk
      constexpr int RPK = 2 * (FK32 + FR32);  // tr reads per kstep
      bf16x4 aT[2][FK32][2], bT[2][FR32][2];  // 2-deep kstep ring
      auto rdk = [&](int kst, int ring) {
#pragma unroll
        for (int f = 0; f < FK32; ++f) {
          const unsigned base =
              aTrM + aOff + (unsigned)((wr * (BK / WR / 16) + f * 2) * KBS +
                                       kst * 4 * TR_MBS) * 2;
          aT[ring][f][0] = ds_tr16(base);
          aT[ring][f][1] = ds_tr16(base + TR_MBS * 2);
        }
#pragma unroll
        for (int f = 0; f < FR32; ++f) {
          const unsigned base =
              bTrM + bOff + (unsigned)((wc * (BR / WC / 16) + f * 2) * KBS +
                                       kst * 4 * TR_MBS) * 2;
          bT[ring][f][0] = ds_tr16(base);
          bT[ring][f][1] = ds_tr16(base + TR_MBS * 2);
        }
      };
      rdk(0, 0);
#pragma unroll
      for (int kst = 0; kst < KST; ++kst) {
        const int ring = kst & 1;
        if (kst + 1 < KST) rdk(kst + 1, ring ^ 1);
        // wait for THIS kstep's reads only; the next kstep's RPK stay
        // in flight under the MFMAs
        if (kst + 1 < KST) {
          if constexpr (RPK == 8)
            asm volatile("s_waitcnt lgkmcnt(8)" ::: "memory");
          else if constexpr (RPK == 6)
            asm volatile("s_waitcnt lgkmcnt(6)" ::: "memory");
          else
            asm volatile("s_waitcnt lgkmcnt(4)" ::: "memory");
        } else {
          asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        }
        __builtin_amdgcn_sched_barrier(0);

#pragma unroll
        for (int fa = 0; fa < FK32; ++fa)
#pragma unroll
          for (int fb = 0; fb < FR32; ++fb) {
            const bf16x8 av = __builtin_shufflevector(
                aT[ring][fa][0], aT[ring][fa][1], 0, 1, 2, 3, 4, 5, 6, 7);
            const bf16x8 bv = __builtin_shufflevector(
                bT[ring][fb][0], bT[ring][fb][1], 0, 1, 2, 3, 4, 5, 6, 7);
            acc32[fa][fb] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                av, bv, acc32[fa][fb], 0, 0, 0);
          }
      }
    } else {
      // tr-read ALL fragments for this chunk into registers (guide T10)
      bf16x4 aT[CH / 32][FK][2], bT[CH / 32][FR][2];
#pragma unroll
      for (int s2 = 0; s2 < CH / 32; ++s2) {
#pragma unroll
        for (int f = 0; f < FK; ++f) {
          const unsigned base = aTr0 + aOff +
                                (unsigned)((wr * FK + f) * KBS +
                                           s2 * 8 * TR_MBS) * 2;
          aT[s2][f][0] = ds_tr16(base);
          aT[s2][f][1] = ds_tr16(base + TR_MBS * 2);
        }
#pragma unroll
        for (int f = 0; f < FR; ++f) {
          const unsigned base = bTr0 + bOff +
                                (unsigned)((wc * FR + f) * KBS +
                                           s2 * 8 * TR_MBS) * 2;
          bT[s2][f][0] = ds_tr16(base);
          bT[s2][f][1] = ds_tr16(base + TR_MBS * 2);
        }
      }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_sched_barrier(0);  // keep MFMAs below the wait
#pragma unroll
      for (int s2 = 0; s2 < CH / 32; ++s2)
#pragma unroll
        for (int fa = 0; fa < FK; ++fa)
#pragma unroll
          for (int fb = 0; fb < FR; ++fb) {
            const bf16x8 av = __builtin_shufflevector(
                aT[s2][fa][0], aT[s2][fa][1], 0, 1, 2, 3, 4, 5, 6, 7);
            const bf16x8 bv = __builtin_shufflevector(
                bT[s2][fb][0], bT[s2][fb][1], 0, 1, 2, 3, 4, 5, 6, 7);
            acc[fa][fb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                av, bv, acc[fa][fb], 0, 0, 0);
          }
    }
    if (next < nChunks) __syncthreads();  // drains DMA + joins waves
    buf ^= 1;
  }

  if constexpr (M32) {
    // ---- 32x32 epilogue: D col = lane&31 (rsc), row = (r&3) + 8*(r>>2)
    //      + 4*(lane>>5) (k) ----
#pragma unroll
    for (int fa = 0; fa < FK32; ++fa) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
        const int k = kt0 + wr * (BK / WR) + fa * 32 + row;
        if (k >= K) continue;
#pragma unroll
        for (int fb = 0; fb < FR32; ++fb) {
          const int rsc = rt0 + wc * (BR / WC) + fb * 32 + (lane & 31);
          if (rsc >= KG) continue;
          const int tap = rsc >> log2Cp;
          const int c = rsc & (Cp - 1);
          if (c >= C) continue;
          const int dy_ = tap / KS, dx_ = tap - (tap / KS) * KS;
          atomicAdd(&dW[(((long)k * C + c) * KS + dy_) * KS + dx_],
                    acc32[fa][fb][r]);
        }
      }
    }
    return;
  }

  // ---- epilogue: scatter-add into NCHW fp32 dW ----
#pragma unroll
  for (int fa = 0; fa < FK; ++fa) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int k = kt0 + (wr * FK + fa) * 16 + lg * 4 + r;
      if (k >= K) continue;
#pragma unroll
      for (int fb = 0; fb < FR; ++fb) {
        int rsc = rt0 + (wc * FR + fb) * 16 + li;
        if (rsc >= KG) continue;
        int tap = rsc >> log2Cp;
        int c = rsc & (Cp - 1);
        if (c >= C) continue;
        int dy = tap / KS, dx = tap - (tap / KS) * KS;
        float v = acc[fa][fb][r];
        atomicAdd(&dW[(((long)k * C + c) * KS + dy) * KS + dx], v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Small-K weight gradient (K <= 4: the 64->3 / 32->3 output convs).
// The MFMA path wastes 13/16 rows of every fragment on Kp padding there;
// a VALU outer-product reduction is ~10x cheaper. Each block owns one
// (tap, m-slice); thread t owns channel octet t % (Cp/8) and accumulates
// acc[K][8] in registers over its m rows, LDS-reduces across the m-rows
// sharing the octet, then atomically adds into the fp32 dW.
// ---------------------------------------------------------------------------

template <int KMAX>
__global__ __launch_bounds__(256, 4) void k_wgrad_smallk(
    const bf16_t* __restrict__ dY,  // (M, Kp)
    const bf16_t* __restrict__ X,   // (N,H,W,Cp)
    float* __restrict__ dW,         // (K, C, KS, KS) fp32
    int N, int H, int W, int Cp, int log2Cp, int Kp, int K, int C, int KS,
    int msplit, unsigned long long mulHW, unsigned long long mulW) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* red = reinterpret_cast<float*>(smem);  // [256][KMAX*8]
  const int PAD = KS / 2;
  const long M = (long)N * H * W;
  const int HW = H * W;
  const int cg = threadIdx.x & (Cp / 8 - 1);   // channel octet
  const int mr = threadIdx.x / (Cp / 8);       // m-row lane
  const int MR = 256 / (Cp / 8);
  const int tap = blockIdx.x;
  const int dy_ = tap / KS, dx_ = tap - dy_ * KS;

  float acc[KMAX][8] = {};
  const long rows = (M + msplit - 1) / msplit;
  const long mstart = (long)blockIdx.y * rows;
  const long mend = min(mstart + rows, M);
  for (long m = mstart + mr; m < mend; m += MR) {
    unsigned n = magic_div((unsigned)m, mulHW);
    unsigned rem = (unsigned)m - n * (unsigned)HW;
    unsigned oy = magic_div(rem, mulW);
    int ox = (int)(rem - oy * (unsigned)W);
    int iy = (int)oy + dy_ - PAD, ix = ox + dx_ - PAD;
    // branch-free halo: clamped load + zero mask — a `continue` here makes
    // hipcc branch around the loads and drain vmcnt per element (guide §5
    // trap 4c), serializing the whole m loop on load latency
    const bool valid = iy >= 0 && iy < H && ix >= 0 && ix < W;
    const int iyc = min(max(iy, 0), H - 1), ixc = min(max(ix, 0), W - 1);
    const bf16x8 xv = *reinterpret_cast<const bf16x8*>(
        X + (((long)((int)n * H + iyc) * W + ixc) << log2Cp) + cg * 8);
    const bf16_t* dyp = dY + m * Kp;
    const float mask = valid ? 1.f : 0.f;
#pragma unroll
    for (int k = 0; k < KMAX; ++k) {
      const float d = mask * bf2f(dyp[k]);
#pragma unroll
      for (int e = 0; e < 8; ++e) acc[k][e] += d * bf2f(xv[e]);
    }
  }
#pragma unroll
  for (int k = 0; k < KMAX; ++k)
#pragma unroll
    for (int e = 0; e < 8; ++e)
      red[(threadIdx.x) * (KMAX * 8) + k * 8 + e] = acc[k][e];
  __syncthreads();
  // parallel tree reduce over the MR m-rows sharing each channel octet
  const int CG = Cp / 8;
  for (int s = MR / 2; s > 0; s >>= 1) {
    if (mr < s) {
#pragma unroll
      for (int k = 0; k < KMAX; ++k)
#pragma unroll
        for (int e = 0; e < 8; ++e)
          red[(mr * CG + cg) * (KMAX * 8) + k * 8 + e] +=
              red[((mr + s) * CG + cg) * (KMAX * 8) + k * 8 + e];
    }
    __syncthreads();
  }
  if (mr == 0) {
    for (int k = 0; k < K && k < KMAX; ++k)
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int c = cg * 8 + e;
        if (c < C)
          atomicAdd(&dW[(((long)k * C + c) * KS + dy_) * KS + dx_],
                    red[cg * (KMAX * 8) + k * 8 + e]);
      }
  }
}

// ---------------------------------------------------------------------------
// tr-read semantics probe: one wave fills a tr image with addr-coded values
// via the SAME writeTiles addressing (tr_addr), tr-reads fragments the same
// way wgrad does, and writes what each (lane, frag, elem) received. The
// host test checks element (lane g*16+i, reg j) == value coded (m=..,col=..).
// ---------------------------------------------------------------------------

__global__ void k_probe_tr(float* __restrict__ out /* [64][2][4] */) {
  __shared__ __attribute__((aligned(16))) bf16_t img[2 * TR_KBS];
  const int lane = threadIdx.x & 63;
  // fill 64 m x 32 col with value m*100 + col (bf16-exact for m<..,col<32)
  for (int m = lane; m < 64; m += 64)
    for (int col = 0; col < 32; ++col)
      img[tr_addr(m, col)] = f2bf((float)(m * 100 + col));
  __syncthreads();
  const int lg = lane >> 4, li = lane & 15;
  const unsigned base0 = (unsigned)(unsigned long long)img +
                         (unsigned)(2 * lg * TR_MBS + li * 4) * 2;
#pragma unroll
  for (int f = 0; f < 2; ++f) {  // two col-blocks (kblk 0, 1)
    const unsigned b = base0 + (unsigned)(f * TR_KBS) * 2;
    bf16x4 lo = ds_tr16(b);
    bf16x4 hi = ds_tr16(b + TR_MBS * 2);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      out[(lane * 2 + f) * 8 + j] = bf2f(lo[j]);
      out[(lane * 2 + f) * 8 + 4 + j] = bf2f(hi[j]);
    }
  }
}

// Raw semantics probe: linear image img[e] = e; lane l reads 8 B at
// byte offset l*8 (elements 4l..4l+3). out[l][j] = delivered element
// index — exposes the hardware's lane/element permutation directly.
__global__ void k_probe_tr_raw(float* __restrict__ out /* [64][4] */) {
  __shared__ __attribute__((aligned(16))) bf16_t img[256];
  const int lane = threadIdx.x & 63;
  for (int e = lane; e < 256; e += 64) img[e] = f2bf((float)e);
  __syncthreads();
  const unsigned addr =
      (unsigned)(unsigned long long)img + (unsigned)lane * 8;
  bf16x4 v = ds_tr16(addr);
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
  for (int j = 0; j < 4; ++j) out[lane * 4 + j] = bf2f(v[j]);
}

at::Tensor probe_tr() {
  auto out = at::zeros({64, 2, 8}, at::TensorOptions()
                                       .dtype(at::kFloat)
                                       .device(at::kCUDA));
  hipStream_t stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(k_probe_tr, dim3(1), dim3(64), 0, stream,
                     out.data_ptr<float>());
  HIP_CHECK_LAST();
  return out;
}

at::Tensor probe_tr_raw() {
  auto out = at::zeros({64, 4}, at::TensorOptions()
                                    .dtype(at::kFloat)
                                    .device(at::kCUDA));
  hipStream_t stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(k_probe_tr_raw, dim3(1), dim3(64), 0, stream,
                     out.data_ptr<float>());
  HIP_CHECK_LAST();
  return out;
}

// ---------------------------------------------------------------------------
// Bias gradient: db[k] = sum_m dY[m][k]
// ---------------------------------------------------------------------------

// Each thread owns one 8-wide k-group and vector-loads bf16x8 rows of dY,
// accumulating 8 fp32 partials in registers; threads covering the same
// k-group at different m-phases reduce through LDS, then each block writes
// a plain fp32 partial slab Part[blockIdx.y][Kp] (no atomics — dB is a few
// cache lines and msplit*K atomics serialize on them; measured 2.2 ms ->
// slabs + the tiny reduce below). Grid: (ceil(ngrp/KGR), msplit).
__global__ void k_bias_grad(const bf16_t* __restrict__ dY,
                            float* __restrict__ Part, long M, int Kp,
                            int msplit) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* red = reinterpret_cast<float*>(smem);  // [256][8] fp32
  const int ngrp = Kp >> 3;                  // 8-wide k-groups in Kp
  const int KGR = min(ngrp, 32);             // k-groups per block
  const int MR = 256 / KGR;                  // m-rows strided per block
  const int kg = threadIdx.x % KGR;          // group id (fastest: coalesced)
  const int mr = threadIdx.x / KGR;
  const int k0 = (blockIdx.x * KGR + kg) * 8;

  float acc[8] = {};
  const long rows = (M + msplit - 1) / msplit;
  const long mstart = (long)blockIdx.y * rows;
  const long mend = min(mstart + rows, M);
  if (k0 < Kp) {
    for (long m = mstart + mr; m < mend; m += MR) {
      const bf16x8 v = *reinterpret_cast<const bf16x8*>(dY + m * Kp + k0);
#pragma unroll
      for (int e = 0; e < 8; ++e) acc[e] += bf2f(v[e]);
    }
  }
#pragma unroll
  for (int e = 0; e < 8; ++e) red[threadIdx.x * 8 + e] = acc[e];
  __syncthreads();
  // tree-reduce over the MR rows that share this thread's k-group
  if (mr == 0 && k0 < Kp) {
#pragma unroll 1
    for (int r = 1; r < MR; ++r)
#pragma unroll
      for (int e = 0; e < 8; ++e) acc[e] += red[(r * KGR + kg) * 8 + e];
    *reinterpret_cast<f32x4*>(&Part[(long)blockIdx.y * Kp + k0]) =
        f32x4{acc[0], acc[1], acc[2], acc[3]};
    *reinterpret_cast<f32x4*>(&Part[(long)blockIdx.y * Kp + k0 + 4]) =
        f32x4{acc[4], acc[5], acc[6], acc[7]};
  }
}

// Reduce the [msplit][Kp] partials into dB (accumulating, like the old
// atomic path: dB may hold a pre-existing gradient). One block per output
// column, 256 threads strided over the msplit rows + LDS tree reduce.
__global__ void k_bias_grad_reduce(const float* __restrict__ Part,
                                   float* __restrict__ dB, int Kp, int K,
                                   int msplit) {
  __shared__ float red[256];
  const int k = blockIdx.x;
  float s = 0.f;
  for (int r = threadIdx.x; r < msplit; r += 256)
    s += Part[(long)r * Kp + k];
  red[threadIdx.x] = s;
  __syncthreads();
  for (int w = 128; w > 0; w >>= 1) {
    if (threadIdx.x < w) red[threadIdx.x] += red[threadIdx.x + w];
    __syncthreads();
  }
  if (threadIdx.x == 0) dB[k] += red[0];
}

// ---------------------------------------------------------------------------
// Weight packing: NCHW fp32 master -> bf16 k-major [Kp][RS*Cp] (fwd) and
// c-major rotated [Cp][RS*Kp] (dgrad).
// ---------------------------------------------------------------------------

__global__ void k_pack_fwd(const float* __restrict__ Wm,  // (K,C,R,S)
                           bf16_t* __restrict__ out,      // [Kp][RS*Cp]
                           int K, int C, int R, int S, int Kp, int Cp) {
  const int KG = R * S * Cp;
  const long total = (long)Kp * KG;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    int k = (int)(idx / KG);
    int rsc = (int)(idx - (long)k * KG);
    int tap = rsc / Cp, c = rsc - (rsc / Cp) * Cp;
    int r = tap / S, s = tap - (tap / S) * S;
    float v = (k < K && c < C) ? Wm[(((long)k * C + c) * R + r) * S + s] : 0.f;
    out[idx] = f2bf(v);
  }
}

__global__ void k_pack_dgrad(const float* __restrict__ Wm,  // (K,C,R,S)
                             bf16_t* __restrict__ out,      // [Cp][RS*Kp]
                             int K, int C, int R, int S, int Kp, int Cp) {
  const int KG = R * S * Kp;
  const long total = (long)Cp * KG;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    int c = (int)(idx / KG);
    int rsk = (int)(idx - (long)c * KG);
    int tap = rsk / Kp, k = rsk - (rsk / Kp) * Kp;
    int rr = tap / S, ss = tap - (tap / S) * S;
    int r = R - 1 - rr, s = S - 1 - ss;  // rotate 180
    float v = (c < C && k < K) ? Wm[(((long)k * C + c) * R + r) * S + s] : 0.f;
    out[idx] = f2bf(v);
  }
}

// Batched repack: one launch packs EVERY layer's fwd + dgrad layouts from
// the fp32 masters (the per-layer k_pack_* launches cost ~5 us each x 36
// per training step under the lazy per-ConvSpec refresh). desc: int64
// [njobs][9] = (Wm, wp, wd, K, C, R, S, Kp, Cp).
__global__ void k_pack_all(const long* __restrict__ desc, int njobs) {
  const long tid0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long nthreads = (long)gridDim.x * blockDim.x;
  for (int j = 0; j < njobs; ++j) {
    const long* d = desc + (long)j * 9;
    const float* Wm = reinterpret_cast<const float*>(d[0]);
    bf16_t* wp = reinterpret_cast<bf16_t*>(d[1]);
    bf16_t* wd = reinterpret_cast<bf16_t*>(d[2]);
    const int K = (int)d[3], C = (int)d[4], R = (int)d[5], S = (int)d[6];
    const int Kp = (int)d[7], Cp = (int)d[8];
    const int KGf = R * S * Cp;
    const long tf = (long)Kp * KGf;
    for (long idx = tid0; idx < tf; idx += nthreads) {
      int k = (int)(idx / KGf);
      int rsc = (int)(idx - (long)k * KGf);
      int tap = rsc / Cp, c = rsc - (rsc / Cp) * Cp;
      int r = tap / S, ss = tap - (tap / S) * S;
      float v =
          (k < K && c < C) ? Wm[(((long)k * C + c) * R + r) * S + ss] : 0.f;
      wp[idx] = f2bf(v);
    }
    const int KGd = R * S * Kp;
    const long td = (long)Cp * KGd;
    for (long idx = tid0; idx < td; idx += nthreads) {
      int c = (int)(idx / KGd);
      int rsk = (int)(idx - (long)c * KGd);
      int tap = rsk / Kp, k = rsk - (rsk / Kp) * Kp;
      int rr = tap / S, sss = tap - (tap / S) * S;
      int r = R - 1 - rr, s2 = S - 1 - sss;  // rotate 180
      float v =
          (c < C && k < K) ? Wm[(((long)k * C + c) * R + r) * S + s2] : 0.f;
      wd[idx] = f2bf(v);
    }
  }
}

void pack_all(const at::Tensor& desc, int64_t njobs) {
  TORCH_CHECK(desc.is_cuda() && desc.dtype() == at::kLong &&
              desc.is_contiguous());
  hipStream_t stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(k_pack_all, dim3(512), dim3(256), 0, stream,
                     desc.data_ptr<long>(), (int)njobs);
  HIP_CHECK_LAST();
}

// ---------------------------------------------------------------------------
// Host-side dispatch
// ---------------------------------------------------------------------------

namespace {

inline int log2i(int v) {
  int l = 0;
  while ((1 << l) < v) ++l;
  TORCH_CHECK((1 << l) == v, "value not a power of two: ", v);
  return l;
}

template <int KS>
void launch_conv_bn(const at::Tensor& x, const at::Tensor& wp,
                    const c10::optional<at::Tensor>& bias, at::Tensor& y,
                    int Klog, int act, hipStream_t stream) {
  const int N = x.size(0), H = x.size(1), W = x.size(2), Cp = x.size(3);
  const int Kp = y.size(3);
  const long M = (long)N * H * W;
  const float* bptr =
      bias.has_value() ? bias->data_ptr<float>() : nullptr;
  const int gx = (int)((M + 127) / 128);
  const int nk = (KS * KS * Cp + 31) / 32;
  // 16 B zero source for glds halo/pad lanes (per-device, created once)
  static thread_local at::Tensor zero16;
  if (!zero16.defined() || zero16.device() != x.device())
    zero16 = at::zeros({8}, x.options());
  const bf16_t* zptr = (const bf16_t*)zero16.data_ptr();

  auto launch = [&](auto bn_const) {
    constexpr int BN = decltype(bn_const)::value;
    const int gy = (Kp + BN - 1) / BN;
    constexpr int NBS = (BN * 4 + 255) / 256;
    constexpr int DKH = (BN <= 32) ? 2 : 1;
    const size_t lds =
        (2 * DKH * 128 * 32 + 2 * DKH * NBS * 256 * 8) * sizeof(bf16_t);
    // Small-M shapes (e.g. VGG 14^2/7^2 layers at bs=16) leave most of the
    // 256 CUs idle; split the K loop across gz slices into fp32 partials,
    // then finalize bias+act+bf16 in a second tiny pass.
    const int base = gx * gy;
    int gz = 1;
    if (base < 320 && nk >= 16) gz = std::min(std::max(1, 512 / base), nk / 4);
    // finalize's unrolled reads tolerate more slabs for the tiniest
    // (grid-limited) shapes; snap >8 to the instantiated 12/16 variants
    gz = std::min(gz, base < 64 ? 16 : 8);
    if (gz > 8) gz = (gz >= 16) ? 16 : 12;
    if (gz > 1) {
      auto y32 = at::empty({gz, M, (long)Kp}, x.options().dtype(at::kFloat));
      const size_t lds5 =
          (5 * 128 * 32 + 5 * NBS * 256 * 8) * sizeof(bf16_t);
      hipLaunchKernelGGL((k_conv_igemm<BN, KS, true>), dim3(gx, gy, gz),
                         dim3(256), lds5, stream,
                         (const bf16_t*)x.data_ptr(),
                         (const bf16_t*)wp.data_ptr(), bptr,
                         (bf16_t*)y.data_ptr(), N, H, W, Cp, log2i(Cp), Kp,
                         Klog, act, zptr, y32.data_ptr<float>());
      const long total = M * Kp;
      const int fb = (int)std::min<long>(512, (total / 4 + 255) / 256);
      auto fin = [&](auto gz_const) {
        constexpr int GZ = decltype(gz_const)::value;
        hipLaunchKernelGGL((k_splitk_finalize<GZ>), dim3(fb), dim3(256), 0,
                           stream, y32.data_ptr<float>(), bptr,
                           (bf16_t*)y.data_ptr(), M, Kp, Klog, act);
      };
      switch (gz) {
        case 2: fin(std::integral_constant<int, 2>{}); break;
        case 3: fin(std::integral_constant<int, 3>{}); break;
        case 4: fin(std::integral_constant<int, 4>{}); break;
        case 5: fin(std::integral_constant<int, 5>{}); break;
        case 6: fin(std::integral_constant<int, 6>{}); break;
        case 7: fin(std::integral_constant<int, 7>{}); break;
        case 12: fin(std::integral_constant<int, 12>{}); break;
        case 16: fin(std::integral_constant<int, 16>{}); break;
        default: fin(std::integral_constant<int, 8>{}); break;
      }
    } else {
      bool done8 = false;
      if constexpr (BN >= 64) {
        static const long m8_thresh = [] {
          // A/B override: the M threshold above which the 8-wave 256-row
          // igemm8 is used instead of the 4-wave 128-row igemm
          const char* e = getenv("WN_IGEMM8_MTHRESH");
          return e ? atol(e) : 16384L;
        }();
        if (M >= m8_thresh) {  // big-M: 8-wave 256-row phase-split kernel
          static const int var = [] {
            const char* e = getenv("WN_IGEMM8_VAR");
            return e ? atoi(e) : 0;
          }();
          static const bool m32 = [] {
            const char* e = getenv("WN_IGEMM8_M32");
            // default OFF: A/B measured a tie on training and -10% at
            // bs=1 1080p inference (the wgrad M32, by contrast, is +5.5%)
            return e != nullptr && atoi(e) != 0;
          }();
          const int gx8 = (int)((M + 255) / 256);
          const int ring = (var == 3) ? 4 : 3;
          const size_t lds8 =
              (size_t)(ring * 256 * 32 + ring * 512 * 8) * sizeof(bf16_t);
          auto l8 = [&](auto var_const) {
            constexpr int V = decltype(var_const)::value;
            hipLaunchKernelGGL((k_conv_igemm8<BN, KS, V>), dim3(gx8, gy),
                               dim3(512), lds8, stream,
                               (const bf16_t*)x.data_ptr(),
                               (const bf16_t*)wp.data_ptr(), bptr,
                               (bf16_t*)y.data_ptr(), N, H, W, Cp,
                               log2i(Cp), Kp, Klog, act, zptr);
          };
          if (m32) {
            hipLaunchKernelGGL((k_conv_igemm8<BN, KS, 0, true>),
                               dim3(gx8, gy), dim3(512), lds8, stream,
                               (const bf16_t*)x.data_ptr(),
                               (const bf16_t*)wp.data_ptr(), bptr,
                               (bf16_t*)y.data_ptr(), N, H, W, Cp,
                               log2i(Cp), Kp, Klog, act, zptr);
          } else {
            switch (var) {
              case 1: l8(std::integral_constant<int, 1>{}); break;
              case 2: l8(std::integral_constant<int, 2>{}); break;
              case 3: l8(std::integral_constant<int, 3>{}); break;
              default: l8(std::integral_constant<int, 0>{});
            }
          }
          done8 = true;
        }
      }
      if (!done8)
        hipLaunchKernelGGL((k_conv_igemm<BN, KS>), dim3(gx, gy), dim3(256),
                           lds, stream, (const bf16_t*)x.data_ptr(),
                           (const bf16_t*)wp.data_ptr(), bptr,
                           (bf16_t*)y.data_ptr(), N, H, W, Cp, log2i(Cp),
                           Kp, Klog, act, zptr);
    }
  };
  if (Kp >= 128)
    launch(std::integral_constant<int, 128>{});
  else if (Kp == 64)
    launch(std::integral_constant<int, 64>{});
  else if (Kp == 32)
    launch(std::integral_constant<int, 32>{});
  else
    launch(std::integral_constant<int, 16>{});
  HIP_CHECK_LAST();
}

}  // namespace

at::Tensor conv2d_fwd(const at::Tensor& x, const at::Tensor& wp,
                      const c10::optional<at::Tensor>& bias, int64_t ks,
                      int64_t Kp, int64_t Klog, int64_t act) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == at::kBFloat16 && x.dim() == 4,
              "x must be CUDA bf16 NHWC");
  TORCH_CHECK(wp.is_cuda() && wp.dtype() == at::kBFloat16);
  TORCH_CHECK(x.is_contiguous() && wp.is_contiguous());
  auto y = at::empty({x.size(0), x.size(1), x.size(2), Kp},
                     x.options());
  hipStream_t stream = at::cuda::getCurrentHIPStream();
  switch (ks) {
    case 1:
      launch_conv_bn<1>(x, wp, bias, y, (int)Klog, (int)act, stream);
      break;
    case 3:
      launch_conv_bn<3>(x, wp, bias, y, (int)Klog, (int)act, stream);
      break;
    case 5:
      launch_conv_bn<5>(x, wp, bias, y, (int)Klog, (int)act, stream);
      break;
    case 7:
      launch_conv_bn<7>(x, wp, bias, y, (int)Klog, (int)act, stream);
      break;
    default:
      TORCH_CHECK(false, "unsupported kernel size ", ks);
  }
  return y;
}

void conv2d_wgrad(const at::Tensor& dy, const at::Tensor& x, at::Tensor& dw,
                  int64_t ks) {
  TORCH_CHECK(dy.is_cuda() && dy.dtype() == at::kBFloat16);
  TORCH_CHECK(x.is_cuda() && x.dtype() == at::kBFloat16);
  TORCH_CHECK(dw.is_cuda() && dw.dtype() == at::kFloat && dw.dim() == 4);
  TORCH_CHECK(dy.is_contiguous() && x.is_contiguous() && dw.is_contiguous());
  const int N = x.size(0), H = x.size(1), W = x.size(2), Cp = x.size(3);
  const int Kp = dy.size(3);
  const int K = dw.size(0), C = dw.size(1);
  TORCH_CHECK(dw.size(2) == ks && dw.size(3) == ks);
  hipStream_t stream0 = at::cuda::getCurrentHIPStream();
  if (K <= 4) {  // output convs (K=3): VALU outer-product path
    const int RS = (int)(ks * ks);
    int msplit = std::max(1, 768 / RS);
    const int MR = 256 / (Cp / 8);
    msplit = (int)std::min<long>(msplit, ((long)N * H * W + MR - 1) / MR);
    hipLaunchKernelGGL((k_wgrad_smallk<4>), dim3(RS, msplit), dim3(256),
                       256 * 32 * sizeof(float), stream0,
                       (const bf16_t*)dy.data_ptr(),
                       (const bf16_t*)x.data_ptr(), dw.data_ptr<float>(), N,
                       H, W, Cp, log2i(Cp), Kp, K, C, (int)ks, msplit,
                       MagicDiv::make((unsigned)(H * W)).mul,
                       MagicDiv::make((unsigned)W).mul);
    HIP_CHECK_LAST();
    return;
  }
  const int KG = (int)(ks * ks) * Cp;
  const int BK = std::min(Kp, 128);
  constexpr int WG_CH = 64;  // m rows per chunk (glds double-buffer budget)
  const int gx = (Kp + BK - 1) / BK, gy = (KG + 127) / 128;
  // split so gx*gy*split fills 256 CUs x ~3 resident blocks
  int split = std::max(1, 640 / std::max(1, gx * gy));
  const long nChunks = ((long)N * H * W + WG_CH - 1) / WG_CH;
  split = (int)std::min<long>(split, nChunks);
  hipStream_t stream = at::cuda::getCurrentHIPStream();
  const int cpk = (WG_CH / 4 * 72 + 16) / 8;  // image chunks per kblk
  const int sa = ((BK / 16) * cpk + 255) / 256;
  const int sb = (8 * cpk + 255) / 256;
  const size_t lds = (size_t)2 * (sa + sb) * 256 * 8 * sizeof(bf16_t);
  static thread_local at::Tensor zero16w;
  if (!zero16w.defined() || zero16w.device() != x.device())
    zero16w = at::zeros({8}, x.options());
  static const bool use_m32 = [] {
    const char* e = getenv("WN_WGRAD_M32");
    return e == nullptr || atoi(e) != 0;  // default ON; 0 = 16x16 path
  }();
  auto launch = [&](auto ks_const, auto bk_const) {
    constexpr int KSV = decltype(ks_const)::value;
    constexpr int BKV = decltype(bk_const)::value;
    constexpr int CHV = 64;
    auto go = [&](auto m32_const) {
      constexpr bool M32V = decltype(m32_const)::value;
      hipLaunchKernelGGL((k_conv_wgrad<KSV, BKV, CHV, M32V>),
                         dim3(gx, gy, split), dim3(256), lds, stream,
                         (const bf16_t*)dy.data_ptr(),
                         (const bf16_t*)x.data_ptr(), dw.data_ptr<float>(),
                         N, H, W, Cp, log2i(Cp), Kp, K, C, split,
                         MagicDiv::make((unsigned)(H * W)).mul,
                         MagicDiv::make((unsigned)W).mul,
                         (const bf16_t*)zero16w.data_ptr());
    };
    if (BKV >= 32 && use_m32)
      go(std::integral_constant<bool, (BKV >= 32)>{});
    else
      go(std::false_type{});
  };
  auto launch_ks = [&](auto ks_const) {
    if (BK == 128)
      launch(ks_const, std::integral_constant<int, 128>{});
    else if (BK == 64)
      launch(ks_const, std::integral_constant<int, 64>{});
    else if (BK == 32)
      launch(ks_const, std::integral_constant<int, 32>{});
    else
      launch(ks_const, std::integral_constant<int, 16>{});
  };
  switch (ks) {
    case 1: launch_ks(std::integral_constant<int, 1>{}); break;
    case 3: launch_ks(std::integral_constant<int, 3>{}); break;
    case 5: launch_ks(std::integral_constant<int, 5>{}); break;
    case 7: launch_ks(std::integral_constant<int, 7>{}); break;
    default: TORCH_CHECK(false, "unsupported kernel size ", ks);
  }
  HIP_CHECK_LAST();
}

void bias_grad(const at::Tensor& dy, at::Tensor& db) {
  TORCH_CHECK(dy.is_cuda() && dy.dtype() == at::kBFloat16);
  TORCH_CHECK(db.is_cuda() && db.dtype() == at::kFloat);
  const int Kp = dy.size(3);
  const long M = dy.numel() / Kp;
  const int K = db.size(0);
  const int ngrp = Kp / 8;
  const int KGR = std::min(ngrp, 32);
  const int gx = (ngrp + KGR - 1) / KGR;
  int msplit = std::max(1, 448 / gx);  // fill the chip; partial slabs
  const int MR = 256 / KGR;
  msplit = (int)std::min<long>(msplit, (M + MR - 1) / MR);
  hipStream_t stream = at::cuda::getCurrentHIPStream();
  auto part = at::empty({msplit, (long)Kp},
                        dy.options().dtype(at::kFloat));
  hipLaunchKernelGGL(k_bias_grad, dim3(gx, msplit), dim3(256),
                     256 * 8 * sizeof(float), stream,
                     (const bf16_t*)dy.data_ptr(), part.data_ptr<float>(),
                     M, Kp, msplit);
  HIP_CHECK_LAST();
  hipLaunchKernelGGL(k_bias_grad_reduce, dim3(K), dim3(256), 0, stream,
                     part.data_ptr<float>(), db.data_ptr<float>(), Kp, K,
                     msplit);
  HIP_CHECK_LAST();
}

at::Tensor pack_weight_fwd(const at::Tensor& w, int64_t Kp, int64_t Cp) {
  TORCH_CHECK(w.is_cuda() && w.dtype() == at::kFloat && w.dim() == 4);
  TORCH_CHECK(w.is_contiguous());
  const int K = w.size(0), C = w.size(1), R = w.size(2), S = w.size(3);
  auto out = at::empty({Kp, (long)R * S * Cp},
                       w.options().dtype(at::kBFloat16));
  const long total = out.numel();
  const int blocks = (int)std::min<long>(2048, (total + 255) / 256);
  hipStream_t stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(k_pack_fwd, dim3(blocks), dim3(256), 0, stream,
                     w.data_ptr<float>(), (bf16_t*)out.data_ptr(), K, C, R, S,
                     (int)Kp, (int)Cp);
  HIP_CHECK_LAST();
  return out;
}

at::Tensor pack_weight_dgrad(const at::Tensor& w, int64_t Kp, int64_t Cp) {
  // output [Cp][RS*Kp]: dgrad conv consumes dY (channels Kp) and produces
  // dX (channels Cp); weights rotated 180 and transposed.
  TORCH_CHECK(w.is_cuda() && w.dtype() == at::kFloat && w.dim() == 4);
  TORCH_CHECK(w.is_contiguous());
  const int K = w.size(0), C = w.size(1), R = w.size(2), S = w.size(3);
  auto out = at::empty({Cp, (long)R * S * Kp},
                       w.options().dtype(at::kBFloat16));
  const long total = out.numel();
  const int blocks = (int)std::min<long>(2048, (total + 255) / 256);
  hipStream_t stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(k_pack_dgrad, dim3(blocks), dim3(256), 0, stream,
                     w.data_ptr<float>(), (bf16_t*)out.data_ptr(), K, C, R, S,
                     (int)Kp, (int)Cp);
  HIP_CHECK_LAST();
  return out;
}
