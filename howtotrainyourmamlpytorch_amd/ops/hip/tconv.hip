// Task-batched 3x3 convolution trio (fwd / dgrad / wgrad) as MFMA
// implicit GEMM on gfx950 — the framework's flagship kernels.
//
// Replaces the reference's per-task F.conv2d calls
// (meta_neural_network_architectures.py:89-97): each task in the meta-batch
// has its OWN weights (fast weights), so ordinary batched conv does not
// apply; MIOpen's grouped path falls back to naive NCHW kernels (measured
// 20 ms/call wgrad — profiles/r01 baseline).  Here:
//
//   fwd   : Y[t]  = im2col(X[t])  @ Wp[t]        M=NS*Ho*Wo, N=Cout, K=9*Cin
//   dgrad : dX[t] = im2col(dY[t]) @ Wp_flip[t]   (SAME kernel, weights
//           repacked flipped+transposed — full correlation identity)
//   wgrad : dW[t] = dY[t]^T @ im2col(X[t])       split-K (v2: operand
//           transposes make both sides k-contiguous for async staging)
//
// Data: bf16 activations/weights (repacked from the fp32 arena), fp32
// accumulate via v_mfma_f32_16x16x32_bf16.  Layout: NHWC (channel-
// innermost = K-contiguous im2col rows).
//
// Two fwd/dgrad generations, both BM=256 / 8 waves / BK=64, bitwise-equal
// outputs (tests pin this):
//   v1: synchronous register staging between two barriers (im2col gather
//       + transposed B writes) — the fallback for Ci % 8 != 0.
//   v2 (default, single-buffer): `global_load_lds` async DMA staging with
//       a padded input (no boundary predicates), a pre-swizzled weight
//       LDS image, per-thread incremental k-state and a zero page for the
//       tails.  Measured: conv1 192->222 TF, omiglot conv1 289->312; the
//       double-buffered 2-phase variant (MAML355_CONV_V2_SBUF=0) LOSES
//       through occupancy (80 KB LDS -> 2 blocks/CU).
// All geometry settled by same-box A/B sweeps — profiles/README.md
// (r1: BM 64->128->256; BK=128 and 2-half-wave variants measured slower
// via LDS-occupancy / issue-pressure cliffs; r2: the pipeline ladder).

#include "common.h"
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

using namespace maml355;

using bf16 = __hip_bfloat16;
typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8;
typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4;

// LDS XOR swizzle (guide T2 / G4): a [rows][64] bf16 tile has a 128-byte row
// stride, so ds_read_b128 of 16 different rows at one column lands 8+ lanes
// per bank.  XOR-ing row bits into the 16-byte-slot bits spreads them; the
// same involution is applied on store and load.  Index in SHORTS; XOR of
// bits 3..5 preserves 16-byte alignment for bf16x8 accesses.
DEVINL int swz64(int row, int col) {
  return (row * 64 + col) ^ ((row & 7) << 3);
}
DEVINL int swz128(int row, int col) {
  return (row * 128 + col) ^ ((row & 7) << 3);
}

// ---------------------------------------------------------------------------
// Weight repack: W [T, F, C, 3, 3] fp32  ->  Wp [T, 9, Ci, Co] bf16
//   fwd  : Wp[t][ky*3+kx][c][f] = W[t][f][c][ky][kx]          (Ci=C, Co=F)
//   dgrad: Wp[t][ky*3+kx][f][c] = W[t][f][c][2-ky][2-kx]      (Ci=F, Co=C)
// ---------------------------------------------------------------------------
__global__ void repack_weight_kernel(const float* __restrict__ w,
                                     bf16* __restrict__ wp,
                                     int T, int F, int C, bool dgrad) {
  const long total = (long)T * F * C * 9;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += grid_stride()) {
    long r = i;
    const int kx = (int)(r % 3); r /= 3;
    const int ky = (int)(r % 3); r /= 3;
    const int c = (int)(r % C); r /= C;
    const int f = (int)(r % F); r /= F;
    const long t = r;
    const float v = w[i];
    long o;
    if (!dgrad) {
      o = (((t * 9 + (long)(ky * 3 + kx)) * C + c) * F) + f;
    } else {
      o = (((t * 9 + (long)((2 - ky) * 3 + (2 - kx))) * F + f) * C) + c;
    }
    wp[o] = __float2bfloat16(v);
  }
}

// fp32 wgrad accumulator [T, nslices, 9C, F] -> dW [T, F, C, 3, 3] fp32;
// K-chunk slices are summed in FIXED order (nslices=1 for the atomic
// fast path; >1 under MAML355_DETERMINISTIC where each K-chunk block
// owns a private slice so the reduction is bitwise reproducible).
__global__ void wgrad_finalize_kernel(const float* __restrict__ acc,
                                      float* __restrict__ dw,
                                      int T, int F, int C, int nslices) {
  const long total = (long)T * F * C * 9;
  const long ssz = (long)9 * C * F;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += grid_stride()) {
    long r = i;
    const int kx = (int)(r % 3); r /= 3;
    const int ky = (int)(r % 3); r /= 3;
    const int c = (int)(r % C); r /= C;
    const int f = (int)(r % F); r /= F;
    const long t = r;
    const long e = ((long)(ky * 3 + kx) * C + c) * F + f;
    float v = 0.f;
    for (int s = 0; s < nslices; ++s) {
      v += acc[(t * nslices + s) * ssz + e];
    }
    dw[i] = v;
  }
}

// ---------------------------------------------------------------------------
// fwd / dgrad kernel.
//   X  [T, NB, H, W, Ci]   bf16 (NB = images per task)
//   Wp [T, 9, Ci, Co]      bf16 (repacked; flipped/transposed for dgrad)
//   bias [T, Co] fp32 or nullptr
//   Y  [T, NB, Ho, Wo, Co] bf16
// pad is the im2col pad (fwd: p; dgrad: 2-p).  stride==1 only (stride-2
// configs use the ATen fallback path).
// ---------------------------------------------------------------------------
#define BM 256
#define BK 64    // K-step: 2 MFMA K-slices per barrier
#define WG_KCHUNK 4096
#define WBK 64   // wgrad K-step (2 MFMA K-slices per barrier)

#define WGN 64   // wgrad output columns per block (4 waves x 16)

__global__ __launch_bounds__(512, 2)
void tconv_mm_kernel(const bf16* __restrict__ X, const bf16* __restrict__ Wp,
                     const float* __restrict__ bias, bf16* __restrict__ Y,
                     float* __restrict__ sums_out,  // [T,2,Co] or nullptr:
                     int T, int NB, int H, int W, int Ci,   // fused BN stats
                     int Ho, int Wo, int Co, int pad) {
  const int t = blockIdx.y;
  const long Mtot = (long)NB * Ho * Wo;
  const long m0 = (long)blockIdx.x * BM;
  const int K9 = 9 * Ci;
  const int ksteps = (K9 + BK - 1) / BK;
  const int ntiles = (Co + 15) / 16;

  __shared__ short lds_a[BM * BK];        // swizzled, see swz64
  __shared__ short lds_bt[64 * BK];        // B^T: [col][k], swizzled
  __shared__ int row_h[BM], row_w[BM], row_n[BM];
  // full k-table hoisted out of the K-loop (9*Ci <= 576 always, since
  // Ci <= 64): one decode per k for the whole block, one fewer barrier
  // per K-step
  __shared__ int ktab_dy[576 + BK], ktab_dx[576 + BK], ktab_c[576 + BK];

  // per-block row decode (once)
  for (int m = threadIdx.x; m < BM; m += blockDim.x) {
    const long mg = m0 + m;
    if (mg < Mtot) {
      const int wo = (int)(mg % Wo);
      const int ho = (int)((mg / Wo) % Ho);
      const int n = (int)(mg / ((long)Wo * Ho));
      row_h[m] = ho; row_w[m] = wo; row_n[m] = n;
    } else {
      row_n[m] = -1;
    }
  }
  for (int k = threadIdx.x; k < ksteps * BK; k += blockDim.x) {
    if (k < K9) {
      const int kyx = k / Ci;
      ktab_c[k] = k - kyx * Ci;
      ktab_dy[k] = kyx / 3;
      ktab_dx[k] = kyx % 3;
    } else {
      ktab_c[k] = -1;
    }
  }

  const bf16* Xt = X + (long)t * NB * H * W * Ci;
  const bf16* Wt = Wp + (long)t * 9 * Ci * Co;

  const int wave = threadIdx.x / WAVE;          // 0..7: 32-row m-subtile
  const int lane = threadIdx.x % WAVE;
  const int fr = lane & 15;                      // fragment row/col
  const int fk = lane >> 4;                      // k-group 0..3

  f32x4 acc[2][4];
#pragma unroll
  for (int hh = 0; hh < 2; ++hh)
#pragma unroll
    for (int i = 0; i < 4; ++i) acc[hh][i] = {0.f, 0.f, 0.f, 0.f};

  __syncthreads();
  for (int ks = 0; ks < ksteps; ++ks) {
    const int k0 = ks * BK;
    // stage A (im2col): slot s -> (m, kk0) with BK/8 slots per row;
    // vector path when the 8-k run stays inside one (ky,kx) slice.
    for (int s = threadIdx.x; s < BM * (BK / 8); s += blockDim.x) {
      const int m = s / (BK / 8);
      const int kk0 = (s % (BK / 8)) * 8;
      const int n = row_n[m];
      const int c0 = ktab_c[k0 + kk0];
      const int c7 = ktab_c[k0 + kk0 + 7];
      if ((Ci % 8 == 0) && c0 >= 0 && c7 == c0 + 7) {
        bf16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
        if (n >= 0) {
          const int h = row_h[m] + ktab_dy[k0 + kk0] - pad;
          const int w = row_w[m] + ktab_dx[k0 + kk0] - pad;
          if (h >= 0 && h < H && w >= 0 && w < W) {
            v = *(const bf16x8*)&(
                (const short*)Xt)[(((long)n * H + h) * W + w) * Ci + c0];
          }
        }
        *(bf16x8*)&lds_a[swz64(m, kk0)] = v;
      } else {
        for (int j = 0; j < 8; ++j) {
          const int kk = kk0 + j;
          const int c = ktab_c[k0 + kk];
          short v = 0;
          if (n >= 0 && c >= 0) {
            const int h = row_h[m] + ktab_dy[k0 + kk] - pad;
            const int w = row_w[m] + ktab_dx[k0 + kk] - pad;
            if (h >= 0 && h < H && w >= 0 && w < W) {
              v = ((const short*)Xt)[(((long)n * H + h) * W + w) * Ci + c];
            }
          }
          lds_a[swz64(m, kk & ~7) + (kk & 7)] = v;
        }
      }
    }
    // stage B^T from the contiguous repacked block at linear k0*Co:
    // slot s -> (kk = s>>3, f0 = (s&7)*8), bf16x8 load + transposed writes
    for (int s = threadIdx.x; s < BK * 8; s += blockDim.x) {
      const int kk = s >> 3;           // 8 f-slots per k row (Co <= 64)
      const int f0 = (s & 7) * 8;
      if ((Co % 8 == 0) && f0 + 8 <= Co) {
        bf16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
        if (k0 + kk < K9) {
          v = *(const bf16x8*)&((const short*)Wt)[(long)(k0 + kk) * Co + f0];
        }
#pragma unroll
        for (int j = 0; j < 8; ++j)
          lds_bt[swz64(f0 + j, kk & ~7) + (kk & 7)] = v[j];
      } else {
        for (int j = 0; j < 8 && f0 + j < 64; ++j) {
          const int f = f0 + j;
          short v = 0;
          if (f < Co && k0 + kk < K9) {
            v = ((const short*)Wt)[(long)(k0 + kk) * Co + f];
          }
          lds_bt[swz64(f, kk & ~7) + (kk & 7)] = v;
        }
      }
    }
    __syncthreads();

    // fragments + MFMA (2 K-slices, 2 row-halves per wave: each B frag
    // feeds two MFMAs)
#pragma unroll
    for (int ksl = 0; ksl < BK / 32; ++ksl) {
      bf16x8 a0 = *(const bf16x8*)&lds_a[swz64(wave * 32 + fr, ksl * 32 + fk * 8)];
      bf16x8 a1 = *(const bf16x8*)&lds_a[swz64(wave * 32 + 16 + fr, ksl * 32 + fk * 8)];
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        if (nt < ntiles) {
          bf16x8 b = *(const bf16x8*)&lds_bt[swz64(nt * 16 + fr, ksl * 32 + fk * 8)];
          acc[0][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b, acc[0][nt], 0, 0, 0);
          acc[1][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b, acc[1][nt], 0, 0, 0);
        }
      }
    }
    __syncthreads();  // all waves done reading before next-step staging
  }

  // epilogue: C/D layout col = lane&15, row = (lane>>4)*4 + j.
  // When sums_out is given, per-column sum/sum-of-squares of the outputs
  // (pre-bf16-rounding, bias included) accumulate for the following BN —
  // removing BN's separate stats pass over the conv output.
  __shared__ float sums_lds[2][64];
  if (sums_out) {
    for (int i = threadIdx.x; i < 128; i += blockDim.x) {
      sums_lds[i >> 6][i & 63] = 0.f;
    }
    __syncthreads();
  }
  bf16* Yt = Y + (long)t * Mtot * Co;
#pragma unroll
  for (int hh = 0; hh < 2; ++hh)
#pragma unroll
  for (int nt = 0; nt < 4; ++nt) {
    if (nt >= ntiles) break;
    const int col = nt * 16 + fr;
    if (col >= Co) continue;
    const float bv = bias ? bias[(long)t * Co + col] : 0.f;
    float ls = 0.f, lq = 0.f;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int row = wave * 32 + hh * 16 + fk * 4 + j;
      const long mg = m0 + row;
      if (mg < Mtot) {
        const float v = acc[hh][nt][j] + bv;
        ((short*)Yt)[mg * Co + col] =
            (short)__bfloat16_as_short(__float2bfloat16(v));
        ls += v;
        lq += v * v;
      }
    }
    if (sums_out) {
      atomicAdd(&sums_lds[0][col], ls);
      atomicAdd(&sums_lds[1][col], lq);
    }
  }
  if (sums_out) {
    __syncthreads();
    for (int c = threadIdx.x; c < Co; c += blockDim.x) {
      atomicAdd(&sums_out[((long)t * 2 + 0) * Co + c], sums_lds[0][c]);
      atomicAdd(&sums_out[((long)t * 2 + 1) * Co + c], sums_lds[1][c]);
    }
  }
}

// ---------------------------------------------------------------------------
// v2 fwd/dgrad: async-staged 2-phase double-buffered schedule.
//
// The v1 kernel stages tiles synchronously through registers between two
// barriers — measured VALU/barrier-bound at 4-8% of bf16 peak (round-1
// profiles).  v2 restructures per the platform guide's pipeline catalog
// (§5.5 T3 minimum-2-phase): while the MFMAs of K-step k run, the loads
// of K-step k+1 are already in flight via `global_load_lds` (async
// global->LDS DMA, 16B/lane, no register round-trip, no VALU staging):
//
//   prologue: STAGE(buf0, ks=0); barrier
//   loop ks:  STAGE(buf^1, ks+1)   // issue async loads, no wait
//             ds_read + MFMA from buf
//             barrier (drains vmcnt -> next tile ready); buf ^= 1
//
// Enablers (all pre-computed on the host side):
//   * input is PADDED ([T,NB,H+2p,W+2p,Ci]) so every im2col address is
//     valid -> no boundary predicates, per-lane source addresses only
//   * weights are repacked into the exact swizzled LDS image
//     ([T,ksteps,64,BK] with the XOR swizzle pre-applied and zeros in
//     the K/Co tails) -> B staging is a linear byte copy
//   * A staging pre-swizzles the SOURCE column slot (j ^= m&7, guide
//     ERRATA 21: global_load_lds writes linearly, so the swizzle must
//     live in the source address; the ds_read side applies the same
//     involution)
//   * K-tail / M-tail lanes read a 16B zero page instead of predicating
// Requires Ci % 8 == 0 (the first conv layer's C in {1,3} keeps v1).
// LDS: 2x(256x64) A + 2x(64x64) B bf16 = 80 KB dynamic -> 2 blocks/CU.
// ---------------------------------------------------------------------------
#define V2_LDS_BYTES (2 * BM * BK * 2 + 2 * 64 * BK * 2)
#define V2_LDS_BYTES_SBUF (BM * BK * 2 + 64 * BK * 2)

// DBUF=true: double-buffered 2-phase (stage ks+1 while computing ks,
// 80 KB LDS -> 2 blocks/CU); DBUF=false: single buffer, stage->compute
// per step (40 KB -> 3-4 blocks/CU; the async DMA still replaces the
// VALU register staging).
template <bool DBUF>
__global__ __launch_bounds__(512, 2)
void tconv_mm_v2_kernel(const bf16* __restrict__ Xp, const bf16* __restrict__ Wimg,
                        const float* __restrict__ bias, const bf16* __restrict__ zpage,
                        bf16* __restrict__ Y, float* __restrict__ sums_out,
                        int T, int NB, int Hp, int Wp, int Ci,
                        int Ho, int Wo, int Co) {
  const int t = blockIdx.y;
  const long Mtot = (long)NB * Ho * Wo;
  const long m0 = (long)blockIdx.x * BM;
  const int K9 = 9 * Ci;
  const int ksteps = (K9 + BK - 1) / BK;
  const int ntiles = (Co + 15) / 16;

  extern __shared__ short smem[];
  // layout: A buffer(s) first, then B buffer(s)
#define LDS_A(buf_) (smem + (DBUF ? (buf_) * (BM * BK) : 0))
#define LDS_B(buf_) \
  (smem + (DBUF ? 2 : 1) * (BM * BK) + (DBUF ? (buf_) * (64 * BK) : 0))

  const bf16* Xt = Xp + (long)t * NB * Hp * Wp * Ci;
  const short* Wt = (const short*)Wimg + (long)t * ksteps * 64 * BK;

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int fr = lane & 15;
  const int fk = lane >> 4;

  // loop-invariant per-thread A-staging descriptors: 4 slots of 16B; slot
  // s -> (row m = s>>3, dest 8-col group j = s&7); source col group is
  // j ^ (m&7) (inverse swizzle on the source side).  The k -> (dy,dx,c)
  // decomposition advances INCREMENTALLY per K-step (adds + compares,
  // no division in the loop).
  int a_rb[4];   // element offset of (n, ho, wo) in the padded input, or -1
  int a_c[4], a_dy[4], a_dx[4], a_koff[4], a_k[4];
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    const int s = q * 512 + threadIdx.x;
    const int m = s >> 3;
    const int k8 = ((s & 7) ^ (m & 7)) * 8;   // k base within the K-step
    a_k[q] = k8;
    const int kyx = k8 / Ci;
    a_c[q] = k8 - kyx * Ci;
    a_dy[q] = kyx / 3;
    a_dx[q] = kyx % 3;
    a_koff[q] = (a_dy[q] * Wp + a_dx[q]) * Ci + a_c[q];
    const long mg = m0 + m;
    if (mg < Mtot) {
      const int wo = (int)(mg % Wo);
      const int ho = (int)((mg / Wo) % Ho);
      const int n = (int)(mg / ((long)Wo * Ho));
      a_rb[q] = ((n * Hp + ho) * Wp + wo) * Ci;
    } else {
      a_rb[q] = -1;
    }
  }
#define V2_ADVANCE()                                                           \
  do {                                                                         \
    _Pragma("unroll")                                                          \
    for (int q = 0; q < 4; ++q) {                                              \
      a_k[q] += BK;                                                            \
      a_c[q] += BK;                                                            \
      while (a_c[q] >= Ci) {                                                   \
        a_c[q] -= Ci;                                                          \
        if (++a_dx[q] == 3) { a_dx[q] = 0; ++a_dy[q]; }                        \
      }                                                                        \
      a_koff[q] = (a_dy[q] * Wp + a_dx[q]) * Ci + a_c[q];                      \
    }                                                                          \
  } while (0)

#define V2_STAGE(ks_, buf_)                                                    \
  do {                                                                         \
    _Pragma("unroll")                                                          \
    for (int q = 0; q < 4; ++q) {                                              \
      const bf16* src = (a_rb[q] < 0 || a_k[q] >= K9)                          \
                            ? zpage                                            \
                            : Xt + a_rb[q] + a_koff[q];                        \
      auto ldst = (__attribute__((address_space(3))) void*)(                   \
          &LDS_A(buf_)[((long)q * 512 + wave * 64) * 8]);                      \
      __builtin_amdgcn_global_load_lds(                                        \
          (const __attribute__((address_space(1))) void*)src, ldst, 16, 0, 0); \
    }                                                                          \
    {                                                                          \
      const short* bsrc = Wt + (long)(ks_)*64 * BK + (long)threadIdx.x * 8;    \
      auto ldst = (__attribute__((address_space(3))) void*)(                   \
          &LDS_B(buf_)[(long)wave * 64 * 8]);                                  \
      __builtin_amdgcn_global_load_lds(                                        \
          (const __attribute__((address_space(1))) void*)bsrc, ldst, 16, 0, 0);\
    }                                                                          \
  } while (0)

  f32x4 acc[2][4];
#pragma unroll
  for (int hh = 0; hh < 2; ++hh)
#pragma unroll
    for (int i = 0; i < 4; ++i) acc[hh][i] = {0.f, 0.f, 0.f, 0.f};

  int buf = 0;
  if (DBUF) {
    V2_STAGE(0, 0);
    __syncthreads();
  }
  for (int ks = 0; ks < ksteps; ++ks) {
    if (DBUF) {
      V2_ADVANCE();
      if (ks + 1 < ksteps) V2_STAGE(ks + 1, buf ^ 1);
    } else {
      V2_STAGE(ks, 0);
      V2_ADVANCE();
      __syncthreads();
    }
    const short* la = LDS_A(buf);
    const short* lb = LDS_B(buf);
#pragma unroll
    for (int ksl = 0; ksl < BK / 32; ++ksl) {
      bf16x8 a0 = *(const bf16x8*)&la[swz64(wave * 32 + fr, ksl * 32 + fk * 8)];
      bf16x8 a1 = *(const bf16x8*)&la[swz64(wave * 32 + 16 + fr, ksl * 32 + fk * 8)];
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        if (nt < ntiles) {
          bf16x8 b = *(const bf16x8*)&lb[swz64(nt * 16 + fr, ksl * 32 + fk * 8)];
          acc[0][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b, acc[0][nt], 0, 0, 0);
          acc[1][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b, acc[1][nt], 0, 0, 0);
        }
      }
    }
    __syncthreads();  // drains vmcnt; all reads of buf done
    if (DBUF) buf ^= 1;
  }
#undef V2_STAGE
#undef V2_ADVANCE

  // epilogue — identical to v1
  __shared__ float sums_lds[2][64];
  if (sums_out) {
    for (int i = threadIdx.x; i < 128; i += blockDim.x) {
      sums_lds[i >> 6][i & 63] = 0.f;
    }
    __syncthreads();
  }
  bf16* Yt = Y + (long)t * Mtot * Co;
#pragma unroll
  for (int hh = 0; hh < 2; ++hh)
#pragma unroll
  for (int nt = 0; nt < 4; ++nt) {
    if (nt >= ntiles) break;
    const int col = nt * 16 + fr;
    if (col >= Co) continue;
    const float bv = bias ? bias[(long)t * Co + col] : 0.f;
    float ls = 0.f, lq = 0.f;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int row = wave * 32 + hh * 16 + fk * 4 + j;
      const long mg = m0 + row;
      if (mg < Mtot) {
        const float v = acc[hh][nt][j] + bv;
        ((short*)Yt)[mg * Co + col] =
            (short)__bfloat16_as_short(__float2bfloat16(v));
        ls += v;
        lq += v * v;
      }
    }
    if (sums_out) {
      atomicAdd(&sums_lds[0][col], ls);
      atomicAdd(&sums_lds[1][col], lq);
    }
  }
  if (sums_out) {
    __syncthreads();
    for (int c = threadIdx.x; c < Co; c += blockDim.x) {
      atomicAdd(&sums_out[((long)t * 2 + 0) * Co + c], sums_lds[0][c]);
      atomicAdd(&sums_out[((long)t * 2 + 1) * Co + c], sums_lds[1][c]);
    }
  }
}

#undef LDS_A
#undef LDS_B

// NHWC zero-pad: [T, NB, H, W, C] -> [T, NB, H+2p, W+2p, C] (C % 8 == 0)
__global__ void pad_nhwc_kernel(const bf16* __restrict__ x, bf16* __restrict__ xp,
                                int T, int NB, int H, int W, int C, int pad) {
  const int Hp = H + 2 * pad, Wp = W + 2 * pad;
  const int c8n = C / 8;
  const long total = (long)T * NB * Hp * Wp * c8n;
  typedef bf16x8 v8;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += grid_stride()) {
    const int c0 = (int)(i % c8n) * 8;
    long r = i / c8n;
    const int wp = (int)(r % Wp); r /= Wp;
    const int hp = (int)(r % Hp); r /= Hp;
    const long nb = r;  // t*NB + nb combined
    const int h = hp - pad, w = wp - pad;
    v8 v = {0, 0, 0, 0, 0, 0, 0, 0};
    if (h >= 0 && h < H && w >= 0 && w < W) {
      v = *(const v8*)&((const short*)x)[((nb * H + h) * W + w) * C + c0];
    }
    *(v8*)&((short*)xp)[((nb * Hp + hp) * Wp + wp) * C + c0] = v;
  }
}

// Weight repack for v2: W [T, F, C, 3, 3] fp32 -> the exact swizzled LDS
// B-image [T, ksteps, 64, BK] bf16 consumed by tconv_mm_v2_kernel
// (zeros in the K9/Co tails; XOR swizzle pre-applied per ERRATA 21).
__global__ void repack_v2_kernel(const float* __restrict__ w,
                                 bf16* __restrict__ out,
                                 int T, int F, int C, int ksteps, bool dgrad) {
  const int cin = dgrad ? F : C;   // GEMM inner-channel count
  const int cout = dgrad ? C : F;  // GEMM output columns
  const int K9 = 9 * cin;
  const long total = (long)T * ksteps * 64 * BK;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += grid_stride()) {
    const int jj = (int)(i & 7);
    const int j = (int)((i >> 3) & 7);
    const int f = (int)((i >> 6) & 63);
    long r = i >> 12;
    const int ks = (int)(r % ksteps);
    const long t = r / ksteps;
    const int k = ks * BK + ((j ^ (f & 7)) * 8) + jj;
    float v = 0.f;
    if (k < K9 && f < cout) {
      const int kyx = k / cin;
      const int ci = k - kyx * cin;
      const int ky = kyx / 3, kx = kyx % 3;
      if (!dgrad) {
        v = w[((((long)t * F + f) * C + ci) * 3 + ky) * 3 + kx];
      } else {
        v = w[((((long)t * F + ci) * C + f) * 3 + (2 - ky)) * 3 + (2 - kx)];
      }
    }
    out[i] = __float2bfloat16(v);
  }
}

// ---------------------------------------------------------------------------
// wgrad v2 support: global transposes that make BOTH wgrad operands
// k-contiguous, so the main kernel stages tiles with linear async
// `global_load_lds` instead of v1's 8-scalar-LDS-write transposes.
//
//   dYT [T, F, Kr]: dY transposed to [f][k'] with each (n, ho) line padded
//     to Wo8 = roundup(Wo, 8) columns (zeros in the pad -> those k
//     contribute nothing), k' = (n*Ho + ho)*Wo8 + wo, Kr = roundup(.., 64).
//   XT  [T, C, NB, Hxp, Wxp]: X transposed channel-major and zero-padded
//     (Hxp = H + 2p, Wxp = Wo8 + 2) so the im2col B-operand read for
//     column n=(c,dy,dx) at position k=(nimg,ho,wo) is the contiguous run
//     XT[c][nimg][ho+dy][wo+dx ..+7] — valid for EVERY C (including the
//     first layer's C in {1,3}, which v1 staged scalar).
// ---------------------------------------------------------------------------

// dY [T,NB,Ho,Wo,F] -> dYT [T,F,Kr]; one block per (t,n,ho) line, tiled
// in 64-wide w-chunks (any Wo).
__global__ void dyt_pad_kernel(const bf16* __restrict__ dy,
                               bf16* __restrict__ dyt,
                               int T, int NB, int Ho, int Wo, int F,
                               int Wo8, long Kr) {
  __shared__ short lds[64 * 72];  // [f][wo] tile, stride 72 (16B-aligned)
  long b = blockIdx.x;
  const int ho = (int)(b % Ho); b /= Ho;
  const int n = (int)(b % NB); b /= NB;
  const int t = (int)b;
  const int f8n = (F + 7) / 8;
  const short* src = (const short*)dy + ((((long)t * NB + n) * Ho + ho) * Wo) * F;
  const long k0 = ((long)n * Ho + ho) * Wo8;
  short* dst = (short*)dyt + (long)t * F * Kr + k0;
  for (int wbase = 0; wbase < Wo8; wbase += 64) {
    const int wlen = min(64, Wo8 - wbase);
    for (int s = threadIdx.x; s < wlen * f8n; s += blockDim.x) {
      const int wl = s / f8n;
      const int wo = wbase + wl;
      const int f0 = (s % f8n) * 8;
      bf16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
      if (wo < Wo && f0 + 8 <= F) {
        v = *(const bf16x8*)&src[(long)wo * F + f0];
      } else if (wo < Wo) {
        for (int j = 0; j < 8 && f0 + j < F; ++j)
          v[j] = src[(long)wo * F + f0 + j];
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) lds[(f0 + j) * 72 + wl] = v[j];
    }
    __syncthreads();
    for (int s = threadIdx.x; s < F * (wlen / 8); s += blockDim.x) {
      const int f = s / (wlen / 8);
      const int w0 = (s % (wlen / 8)) * 8;
      *(bf16x8*)&dst[(long)f * Kr + wbase + w0] =
          *(const bf16x8*)&lds[f * 72 + w0];
    }
    __syncthreads();
  }
}

// zero dYT's K' .. Kr tail (< 64 columns per row)
__global__ void dyt_tail_kernel(bf16* __restrict__ dyt, int T, int F,
                                long Kprime, long Kr) {
  const long tail = Kr - Kprime;
  const long total = (long)T * F * tail;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += grid_stride()) {
    const long c = i % tail;
    const long r = i / tail;
    ((short*)dyt)[r * Kr + Kprime + c] = 0;
  }
}

// X [T,NB,H,W,C] -> XT [T,C,NB,Hxp,Wxp]; block per (t, n, hxp, w-tile).
__global__ void xt_pad_kernel(const bf16* __restrict__ x, bf16* __restrict__ xt,
                              int T, int NB, int H, int W, int C,
                              int Hxp, int Wxp, int pad) {
  __shared__ short lds[64 * 72];  // [c][wx] tile
  long b = blockIdx.x;
  const int wt = (int)(b % ((Wxp + 63) / 64)); b /= (Wxp + 63) / 64;
  const int hxp = (int)(b % Hxp); b /= Hxp;
  const int n = (int)(b % NB); b /= NB;
  const int t = (int)b;
  const int h = hxp - pad;
  const int w0 = wt * 64;
  const int c8n = (C + 7) / 8;
  const short* src = (const short*)x + (((long)t * NB + n) * H) * W * C;
  for (int s = threadIdx.x; s < 64 * c8n; s += blockDim.x) {
    const int wl = s / c8n;
    const int c0 = (s % c8n) * 8;
    const int w = w0 + wl - pad;
    bf16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
    if (h >= 0 && h < H && w >= 0 && w < W) {
      if (c0 + 8 <= C) {
        v = *(const bf16x8*)&src[((long)h * W + w) * C + c0];
      } else {
        for (int j = 0; j < 8 && c0 + j < C; ++j)
          v[j] = src[((long)h * W + w) * C + c0 + j];
      }
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) lds[(c0 + j) * 72 + wl] = v[j];
  }
  __syncthreads();
  short* dst = (short*)xt + ((((long)t * C) * NB + n) * Hxp + hxp) * Wxp;
  for (int s = threadIdx.x; s < C * 8; s += blockDim.x) {
    const int c = s / 8;
    const int wg = (s % 8) * 8;
    const int wx = w0 + wg;
    if (wx + 8 <= Wxp) {
      *(bf16x8*)&dst[((long)c * NB * Hxp) * Wxp + wx] =
          *(const bf16x8*)&lds[c * 72 + wg];
    } else {
      for (int j = 0; j < 8 && wx + j < Wxp; ++j)
        dst[((long)c * NB * Hxp) * Wxp + wx + j] = lds[c * 72 + wg + j];
    }
  }
}

// async-staged wgrad main kernel: A = dYT rows (f), B = im2col columns
// from XT; 4 waves, double-buffered tiles, 2-phase schedule.  NSUB = how
// many 64-wide n-column blocks this block owns (1 or 2): NSUB=2 doubles
// the MFMA work amortizing each barrier at 2x the B-tile LDS.
template <int NSUB, bool DET>
__global__ __launch_bounds__(256, 4)
void tconv_wgrad_v2_kernel(const bf16* __restrict__ dYT,
                           const bf16* __restrict__ XT,
                           const bf16* __restrict__ zpage,
                           float* __restrict__ dWacc,  // [T, nsl, 9C, F]
                           float* __restrict__ dBacc,  // [T, nsl, F] or null
                           int T, int NB, int Ho, int Wo8, int Hxp, int Wxp,
                           int C, int F, long Kr, long Kprime, int kchunk) {
  const int t = blockIdx.z;
  const int n0 = blockIdx.x * (64 * NSUB);
  const int N9 = 9 * C;
  const long kc0 = (long)blockIdx.y * kchunk;
  const long kc_end = min(kc0 + (long)kchunk, Kr);
  const int mtiles = (F + 15) / 16;
  const long imgS = (long)NB * Hxp * Wxp;

  __shared__ short lds_at[2][64 * WBK];
  __shared__ short lds_bt[2][64 * NSUB * WBK];

  const bf16* dYTt = dYT + (long)t * F * Kr;
  const bf16* XTt = XT + (long)t * C * imgS;

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int fr = lane & 15;
  const int fk = lane >> 4;

  // ---- per-thread staging descriptors (2 A slots + 2 B slots) ----
  // A slot s: row f = s>>3, dest col group j = s&7, src group j^(f&7)
  int a_f[2], a_k8[2];
#pragma unroll
  for (int q = 0; q < 2; ++q) {
    const int s = q * 256 + threadIdx.x;
    a_f[q] = s >> 3;
    a_k8[q] = ((s & 7) ^ ((s >> 3) & 7)) * 8;
  }
  // B slot s: col n = s>>3 (global n0+n), src k group (s&7)^(n&7);
  // incremental k -> (image row offset, wo8) state per slot so the main
  // loop advances with adds/compares only (no division)
  long b_imgoff[2 * NSUB];  // c*imgS + ((nimg*Hxp)+ho+dy)*Wxp + dx
  int b_wo8[2 * NSUB], b_ho[2 * NSUB];
  long b_k[2 * NSUB];
  bool b_valid[2 * NSUB];
#pragma unroll
  for (int q = 0; q < 2 * NSUB; ++q) {
    const int s = q * 256 + threadIdx.x;
    const int n = s >> 3;
    const int ng = n0 + n;
    b_valid[q] = ng < N9;
    const int kyx = b_valid[q] ? ng / C : 0;
    const int c = ng - kyx * C;
    b_k[q] = kc0 + ((s & 7) ^ (n & 7)) * 8;
    const int wo8 = (int)(b_k[q] % Wo8);
    const long line = b_k[q] / Wo8;
    const int ho = (int)(line % Ho);
    const int nimg = (int)(line / Ho);
    b_wo8[q] = wo8;
    b_ho[q] = ho;
    b_imgoff[q] = ((long)c * NB + nimg) * Hxp * Wxp +
                  (long)(ho + kyx / 3) * Wxp + kyx % 3;
  }

#define WG2_STAGE(k0_, buf_)                                                   \
  do {                                                                         \
    _Pragma("unroll")                                                          \
    for (int q = 0; q < 2; ++q) {                                              \
      const bf16* src = (a_f[q] < F)                                           \
                            ? dYTt + (long)a_f[q] * Kr + (k0_) + a_k8[q]       \
                            : zpage;                                           \
      auto ldst = (__attribute__((address_space(3))) void*)(                   \
          &lds_at[buf_][((long)q * 256 + wave * 64) * 8]);                     \
      __builtin_amdgcn_global_load_lds(                                        \
          (const __attribute__((address_space(1))) void*)src, ldst, 16, 0, 0); \
    }                                                                          \
    _Pragma("unroll")                                                          \
    for (int q = 0; q < 2 * NSUB; ++q) {                                       \
      const bf16* src = (b_valid[q] && b_k[q] < Kprime)                        \
                            ? XTt + b_imgoff[q] + b_wo8[q]                     \
                            : zpage;                                           \
      auto ldst = (__attribute__((address_space(3))) void*)(                   \
          &lds_bt[buf_][((long)q * 256 + wave * 64) * 8]);                     \
      __builtin_amdgcn_global_load_lds(                                        \
          (const __attribute__((address_space(1))) void*)src, ldst, 16, 0, 0); \
    }                                                                          \
  } while (0)

  // advance the B k-state by one K-step (WBK columns)
#define WG2_ADVANCE()                                                          \
  do {                                                                         \
    _Pragma("unroll")                                                          \
    for (int q = 0; q < 2 * NSUB; ++q) {                                       \
      b_k[q] += WBK;                                                           \
      b_wo8[q] += WBK;                                                         \
      while (b_wo8[q] >= Wo8) {                                                \
        b_wo8[q] -= Wo8;                                                       \
        ++b_ho[q];                                                             \
        b_imgoff[q] += Wxp;                                                    \
        if (b_ho[q] == Ho) {                                                   \
          b_ho[q] = 0;                                                         \
          b_imgoff[q] += (long)(Hxp - Ho) * Wxp;                               \
        }                                                                      \
      }                                                                        \
    }                                                                          \
  } while (0)

  f32x4 acc[NSUB][4];
#pragma unroll
  for (int sb = 0; sb < NSUB; ++sb)
#pragma unroll
    for (int i = 0; i < 4; ++i) acc[sb][i] = {0.f, 0.f, 0.f, 0.f};
  const bool do_bias = (dBacc != nullptr) && (blockIdx.x == 0);
  float db_acc = 0.f;
  const int db_f = threadIdx.x & 63;
  const int db_q = threadIdx.x >> 6;

  WG2_STAGE(kc0, 0);
  __syncthreads();
  int buf = 0;
  for (long k0 = kc0; k0 < kc_end; k0 += WBK) {
    WG2_ADVANCE();
    if (k0 + WBK < kc_end) WG2_STAGE(k0 + WBK, buf ^ 1);
    const short* la = lds_at[buf];
    const short* lb = lds_bt[buf];
#pragma unroll
    for (int ks = 0; ks < WBK / 32; ++ks) {
#pragma unroll
      for (int sb = 0; sb < NSUB; ++sb) {
        // wave's n-subtile within B rows: (sb*4 + wave)*16
        bf16x8 bfrag = *(const bf16x8*)&lb[swz64((sb * 4 + wave) * 16 + fr,
                                                 ks * 32 + fk * 8)];
#pragma unroll
        for (int mt = 0; mt < 4; ++mt) {
          if (mt < mtiles) {
            bf16x8 afrag = *(const bf16x8*)&la[swz64(mt * 16 + fr, ks * 32 + fk * 8)];
            acc[sb][mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[sb][mt], 0, 0, 0);
          }
        }
      }
    }
    if (do_bias && db_f < F) {
#pragma unroll
      for (int kk = db_q * 16; kk < db_q * 16 + 16; ++kk) {
        db_acc += __bfloat162float(__hip_bfloat16(__hip_bfloat16_raw{
            (unsigned short)la[swz64(db_f, kk & ~7) + (kk & 7)]}));
      }
    }
    __syncthreads();
    buf ^= 1;
  }
#undef WG2_STAGE
#undef WG2_ADVANCE

  if (do_bias) {
    __shared__ float db_lds[4][64];
    db_lds[db_q][db_f] = db_acc;
    __syncthreads();
    if (db_q == 0 && db_f < F) {
      const float v = db_lds[0][db_f] + db_lds[1][db_f] + db_lds[2][db_f] +
                      db_lds[3][db_f];
      if (DET) {
        dBacc[((long)t * gridDim.y + blockIdx.y) * F + db_f] = v;
      } else {
        atomicAdd(&dBacc[(long)t * F + db_f], v);
      }
    }
  }

  float* dWt = DET ? dWacc + ((long)t * gridDim.y + blockIdx.y) * N9 * F
                   : dWacc + (long)t * N9 * F;
#pragma unroll
  for (int sb = 0; sb < NSUB; ++sb)
#pragma unroll
  for (int mt = 0; mt < 4; ++mt) {
    if (mt >= mtiles) continue;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int f = mt * 16 + fk * 4 + j;
      const int n = n0 + (sb * 4 + wave) * 16 + fr;
      if (f < F && n < N9) {
        if (DET) {
          dWt[(long)n * F + f] = acc[sb][mt][j];
        } else {
          atomicAdd(&dWt[(long)n * F + f], acc[sb][mt][j]);
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// wgrad kernel: dWacc[t][n=9C][f] += sum_k dY[t,k,f] * im2col(X)[t,k,n]
// Block: 256 thr = 4 waves; wave w owns n-subtile w (16 cols), iterates
// m-tiles over F.  K-chunked grid with fp32 atomicAdd.
//   dY [T, NB, Ho, Wo, F] bf16 ; X [T, NB, H, W, C] bf16
// ---------------------------------------------------------------------------

// DET=true: each (t, K-chunk) block writes a PRIVATE accumulator slice
// (plain stores, no atomics) — deterministic; DET=false: fp32 atomicAdd
// into the shared per-task accumulator (fast path).
template <bool DET>
__global__ __launch_bounds__(256, 2)
void tconv_wgrad_kernel(const bf16* __restrict__ dY, const bf16* __restrict__ X,
                        float* __restrict__ dWacc,  // [T, nsl, 9C, F]
                        float* __restrict__ dBacc,  // [T, nsl, F] or nullptr
                        int T, int NB, int H, int W, int C,
                        int Ho, int Wo, int F, int pad, int kchunk) {
  const int t = blockIdx.z;
  const int n0 = blockIdx.x * WGN;         // column block within 9C
  const int N9 = 9 * C;
  const long Ktot = (long)NB * Ho * Wo;
  const long kchunk0 = (long)blockIdx.y * kchunk;
  const long kchunk_end = min(kchunk0 + (long)kchunk, Ktot);
  const int mtiles = (F + 15) / 16;

  __shared__ short lds_at[64 * WBK];   // dY^T tile: [f][k], swizzled
  __shared__ short lds_bt[WGN * WBK];  // im2col^T tile: [n][k], swizzled
  __shared__ int ntab_dy[WGN], ntab_dx[WGN], ntab_c[WGN];
  __shared__ int ktab_n[WBK], ktab_h[WBK], ktab_w[WBK];  // k -> image pos

  // n-table (once): n -> (dy, dx, c)
  for (int e = threadIdx.x; e < WGN; e += blockDim.x) {
    const int n = n0 + e;
    if (n < N9) {
      const int kyx = n / C;
      ntab_c[e] = n - kyx * C;
      ntab_dy[e] = kyx / 3;
      ntab_dx[e] = kyx % 3;
    } else {
      ntab_c[e] = -1;
    }
  }

  const bf16* dYt = dY + (long)t * Ktot * F;
  const bf16* Xt = X + (long)t * NB * H * W * C;

  const int wave = threadIdx.x / WAVE;      // n-subtile
  const int lane = threadIdx.x % WAVE;
  const int fr = lane & 15;
  const int fk = lane >> 4;

  f32x4 acc[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};
  // fused bias-grad: only the n0==0 column block accumulates (the dY tile
  // is identical across column blocks)
  const bool do_bias = (dBacc != nullptr) && (blockIdx.x == 0);
  float db_acc = 0.f;
  const int db_f = threadIdx.x & 63;
  const int db_q = threadIdx.x >> 6;        // k-range slice (8 x 8 with 512 thr)

  for (long k0 = kchunk0; k0 < kchunk_end; k0 += WBK) {
    __syncthreads();
    // k-position table: one decode per k instead of per staged element
    if (threadIdx.x < WBK) {
      const long k = k0 + threadIdx.x;
      if (k < kchunk_end) {
        const int wo = (int)(k % Wo);
        const long r = k / Wo;
        ktab_w[threadIdx.x] = wo;
        ktab_h[threadIdx.x] = (int)(r % Ho);
        ktab_n[threadIdx.x] = (int)(r / Ho);
      } else {
        ktab_n[threadIdx.x] = -1;
      }
    }
    __syncthreads();
    // stage dY^T, vectorized: slot s -> (kk = s>>3, f0 = (s&7)*8); one
    // bf16x8 load of 8 consecutive f, 8 scalar LDS writes (transpose).
    // F is always a multiple of 16 here (conv out-channels 48/64), and
    // rows beyond F are never read by the fragments (mtiles*16 == F).
    for (int s = threadIdx.x; s < WBK * 8; s += blockDim.x) {
      const int kk = s >> 3;
      const int f0 = (s & 7) * 8;
      if (f0 >= F) continue;
      bf16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
      if (ktab_n[kk] >= 0) {
        v = *(const bf16x8*)&((const short*)dYt)[(k0 + kk) * F + f0];
      }
#pragma unroll
      for (int j = 0; j < 8; ++j)
        lds_at[swz64(f0 + j, kk & ~7) + (kk & 7)] = v[j];
    }
    // stage im2col^T: vector fast path when the 8-column run stays inside
    // one (ky,kx) slice (contiguous c, 16B-aligned when C % 8 == 0);
    // scalar fallback otherwise (first-layer C in {1,3}).
    for (int s = threadIdx.x; s < WBK * (WGN / 8); s += blockDim.x) {
      const int kk = s / (WGN / 8);
      const int n8 = (s % (WGN / 8)) * 8;
      const int nimg = ktab_n[kk];
      const int c0 = ntab_c[n8];
      const int c7 = ntab_c[n8 + 7];
      const bool vec_ok = (C % 8 == 0) && c0 >= 0 && c7 == c0 + 7;
      if (vec_ok) {
        bf16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
        if (nimg >= 0) {
          const int h = ktab_h[kk] + ntab_dy[n8] - pad;
          const int w = ktab_w[kk] + ntab_dx[n8] - pad;
          if (h >= 0 && h < H && w >= 0 && w < W) {
            v = *(const bf16x8*)&(
                (const short*)Xt)[(((long)nimg * H + h) * W + w) * C + c0];
          }
        }
#pragma unroll
        for (int j = 0; j < 8; ++j)
          lds_bt[swz64(n8 + j, kk & ~7) + (kk & 7)] = v[j];
      } else {
        for (int j = 0; j < 8; ++j) {
          const int ncol = n8 + j;
          const int c = ntab_c[ncol];
          short v = 0;
          if (nimg >= 0 && c >= 0) {
            const int h = ktab_h[kk] + ntab_dy[ncol] - pad;
            const int w = ktab_w[kk] + ntab_dx[ncol] - pad;
            if (h >= 0 && h < H && w >= 0 && w < W) {
              v = ((const short*)Xt)[(((long)nimg * H + h) * W + w) * C + c];
            }
          }
          lds_bt[swz64(ncol, kk & ~7) + (kk & 7)] = v;
        }
      }
    }
    __syncthreads();

#pragma unroll
    for (int ks = 0; ks < WBK / 32; ++ks) {
      bf16x8 b = *(const bf16x8*)&lds_bt[swz64(wave * 16 + fr, ks * 32 + fk * 8)];
#pragma unroll
      for (int mt = 0; mt < 4; ++mt) {
        if (mt < mtiles) {
          bf16x8 a = *(const bf16x8*)&lds_at[swz64(mt * 16 + fr, ks * 32 + fk * 8)];
          acc[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[mt], 0, 0, 0);
        }
      }
    }
    if (do_bias && db_f < F) {
#pragma unroll
      for (int kk = db_q * 16; kk < db_q * 16 + 16; ++kk) {
        db_acc += __bfloat162float(__hip_bfloat16(__hip_bfloat16_raw{
            (unsigned short)lds_at[swz64(db_f, kk & ~7) + (kk & 7)]}));
      }
    }
  }
  if (do_bias) {
    __shared__ float db_lds[4][64];
    db_lds[db_q][db_f] = db_acc;
    __syncthreads();
    if (db_q == 0 && db_f < F) {
      const float v = db_lds[0][db_f] + db_lds[1][db_f] + db_lds[2][db_f] +
                      db_lds[3][db_f];
      if (DET) {
        dBacc[((long)t * gridDim.y + blockIdx.y) * F + db_f] = v;
      } else {
        atomicAdd(&dBacc[(long)t * F + db_f], v);
      }
    }
  }

  // accumulate: C/D col = lane&15 (n within wave tile), row = fk*4+j (f)
  // (n-columns are disjoint across blockIdx.x, so under DET the private
  // (t, K-chunk) slice needs no atomics at all)
  float* dWt = DET ? dWacc + ((long)t * gridDim.y + blockIdx.y) * N9 * F
                   : dWacc + (long)t * N9 * F;
#pragma unroll
  for (int mt = 0; mt < 4; ++mt) {
    if (mt >= mtiles) break;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int f = mt * 16 + fk * 4 + j;
      const int n = n0 + wave * 16 + fr;
      if (f < F && n < N9) {
        if (DET) {
          dWt[(long)n * F + f] = acc[mt][j];
        } else {
          atomicAdd(&dWt[(long)n * F + f], acc[mt][j]);
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// MFMA layout probe: D[16x16] = A[16x32] x B[32x16], single wave.
// Used by the GPU tests to pin the fragment layout against torch.matmul
// with asymmetric random matrices (guide rule G9).
// ---------------------------------------------------------------------------
__global__ void mfma_probe_kernel(const bf16* __restrict__ A,
                                  const bf16* __restrict__ B,
                                  float* __restrict__ D) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int fr = lane & 15;
  const int fk = lane >> 4;
  bf16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = ((const short*)A)[fr * 32 + fk * 8 + j];        // A[row][k]
    b[j] = ((const short*)B)[(fk * 8 + j) * 16 + fr];      // B[k][col]
  }
  f32x4 d = {0.f, 0.f, 0.f, 0.f};
  d = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, d, 0, 0, 0);
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    D[(fk * 4 + j) * 16 + fr] = d[j];                      // D[row][col]
  }
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------
namespace {
int ew_grid2(long total, int threads) {
  long blocks = (total + threads - 1) / threads;
  return (int)std::min<long>(blocks, 4096);
}
}  // namespace

torch::Tensor tconv_repack(torch::Tensor w, bool dgrad) {
  TORCH_CHECK(w.is_cuda() && w.dim() == 5 && w.size(3) == 3 && w.size(4) == 3);
  auto wc = w.contiguous().to(torch::kFloat32);
  const int T = (int)w.size(0), F = (int)w.size(1), C = (int)w.size(2);
  auto wp = torch::empty({T, 9, dgrad ? F : C, dgrad ? C : F},
                         w.options().dtype(torch::kBFloat16));
  auto stream = at::cuda::getCurrentCUDAStream();
  const long total = (long)T * F * C * 9;
  hipLaunchKernelGGL(repack_weight_kernel, dim3(ew_grid2(total, 256)), dim3(256),
                     0, stream.stream(), wc.data_ptr<float>(),
                     reinterpret_cast<bf16*>(wp.data_ptr()), T, F, C, dgrad);
  return wp;
}

// x [T, NB, H, W, Ci] bf16 ; wp [T, 9, Ci, Co] bf16 ; bias [T, Co] fp32 / undef
// returns {y, sums[T,2,Co]} — sums populated only when with_stats
std::vector<torch::Tensor> tconv_mm(torch::Tensor x, torch::Tensor wp,
                                    c10::optional<torch::Tensor> bias, long pad,
                                    long Ho, long Wo, bool with_stats) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 5 && x.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "tconv_mm needs bf16");
  const int T = (int)x.size(0), NB = (int)x.size(1), H = (int)x.size(2),
            W = (int)x.size(3), Ci = (int)x.size(4);
  const int Co = (int)wp.size(3);
  TORCH_CHECK(wp.size(2) == Ci && Co <= 64, "Co must be <= 64");
  auto y = torch::empty({T, NB, Ho, Wo, Co}, x.options());
  auto sums = with_stats
                  ? torch::zeros({T, 2, Co}, x.options().dtype(torch::kFloat32))
                  : torch::empty({0}, x.options().dtype(torch::kFloat32));
  const long Mtot = (long)NB * Ho * Wo;
  dim3 grid((unsigned)((Mtot + BM - 1) / BM), T);
  auto stream = at::cuda::getCurrentCUDAStream();
  const int mm_threads = 512;  // 8 waves: one 16-row m-subtile each
  const float* bptr = nullptr;
  torch::Tensor bc;
  if (bias.has_value()) {
    bc = bias->contiguous().to(torch::kFloat32);
    bptr = bc.data_ptr<float>();
  }
  hipLaunchKernelGGL(tconv_mm_kernel, grid, dim3(mm_threads), 0, stream.stream(),
                     reinterpret_cast<const bf16*>(x.data_ptr()),
                     reinterpret_cast<const bf16*>(wp.data_ptr()), bptr,
                     reinterpret_cast<bf16*>(y.data_ptr()),
                     with_stats ? sums.data_ptr<float>() : nullptr,
                     T, NB, H, W, Ci, (int)Ho, (int)Wo, Co, (int)pad);
  return {y, sums};
}

torch::Tensor tconv_repack_v2(torch::Tensor w, bool dgrad) {
  TORCH_CHECK(w.is_cuda() && w.dim() == 5 && w.size(3) == 3 && w.size(4) == 3);
  auto wc = w.contiguous().to(torch::kFloat32);
  const int T = (int)w.size(0), F = (int)w.size(1), C = (int)w.size(2);
  const int cin = dgrad ? F : C;
  const int ksteps = (9 * cin + BK - 1) / BK;
  auto out = torch::empty({T, ksteps, 64, BK},
                          w.options().dtype(torch::kBFloat16));
  auto stream = at::cuda::getCurrentCUDAStream();
  const long total = (long)T * ksteps * 64 * BK;
  hipLaunchKernelGGL(repack_v2_kernel, dim3(ew_grid2(total, 256)), dim3(256), 0,
                     stream.stream(), wc.data_ptr<float>(),
                     reinterpret_cast<bf16*>(out.data_ptr()), T, F, C, ksteps,
                     dgrad);
  return out;
}

// v2 conv: x UNPADDED [T, NB, H, W, Ci] bf16 (Ci % 8 == 0), wimg from
// tconv_repack_v2 (Co zero-padded to 64 rows, so the caller passes the
// true Co); pads internally; returns {y, sums}
std::vector<torch::Tensor> tconv_mm_v2(torch::Tensor x, torch::Tensor wimg,
                                       c10::optional<torch::Tensor> bias,
                                       long pad, long Ho, long Wo, long Co,
                                       bool with_stats) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 5 && x.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "tconv_mm_v2 needs bf16");
  const int T = (int)x.size(0), NB = (int)x.size(1), H = (int)x.size(2),
            W = (int)x.size(3), Ci = (int)x.size(4);
  TORCH_CHECK(Ci % 8 == 0 && Co <= 64);
  auto stream = at::cuda::getCurrentCUDAStream();
  // pad (p == 0 -> use x directly: every im2col address is already valid)
  torch::Tensor xp = x;
  int Hp = H, Wp = W;
  if (pad > 0) {
    Hp = H + 2 * (int)pad;
    Wp = W + 2 * (int)pad;
    xp = torch::empty({T, NB, Hp, Wp, Ci}, x.options());
    const long ptotal = (long)T * NB * Hp * Wp * (Ci / 8);
    hipLaunchKernelGGL(pad_nhwc_kernel, dim3(ew_grid2(ptotal, 256)), dim3(256),
                       0, stream.stream(),
                       reinterpret_cast<const bf16*>(x.data_ptr()),
                       reinterpret_cast<bf16*>(xp.data_ptr()),
                       T, NB, H, W, Ci, (int)pad);
  }
  auto y = torch::empty({T, NB, Ho, Wo, Co}, x.options());
  auto sums = with_stats
                  ? torch::zeros({T, 2, Co}, x.options().dtype(torch::kFloat32))
                  : torch::empty({0}, x.options().dtype(torch::kFloat32));
  auto zpage_cache = torch::zeros({16}, x.options());
  const float* bptr = nullptr;
  torch::Tensor bc;
  if (bias.has_value()) {
    bc = bias->contiguous().to(torch::kFloat32);
    bptr = bc.data_ptr<float>();
  }
  const long Mtot = (long)NB * Ho * Wo;
  dim3 grid((unsigned)((Mtot + BM - 1) / BM), T);
  // single-buffer is the measured default (conv1 192->222 TF vs v1;
  // the 80 KB double-buffer drops occupancy 3->2 blocks/CU and loses)
  const char* sbuf_env = getenv("MAML355_CONV_V2_SBUF");
  const bool sbuf = !(sbuf_env && sbuf_env[0] == '0');
  static bool attr_set = false;
  if (!attr_set) {
    hipFuncSetAttribute((const void*)tconv_mm_v2_kernel<true>,
                        hipFuncAttributeMaxDynamicSharedMemorySize,
                        V2_LDS_BYTES);
    hipFuncSetAttribute((const void*)tconv_mm_v2_kernel<false>,
                        hipFuncAttributeMaxDynamicSharedMemorySize,
                        V2_LDS_BYTES_SBUF);
    attr_set = true;
  }
#define LAUNCH_MMV2(DB_, BYTES_)                                               \
  hipLaunchKernelGGL((tconv_mm_v2_kernel<DB_>), grid, dim3(512), BYTES_,       \
                     stream.stream(),                                          \
                     reinterpret_cast<const bf16*>(xp.data_ptr()),             \
                     reinterpret_cast<const bf16*>(wimg.data_ptr()), bptr,     \
                     reinterpret_cast<const bf16*>(zpage_cache.data_ptr()),    \
                     reinterpret_cast<bf16*>(y.data_ptr()),                    \
                     with_stats ? sums.data_ptr<float>() : nullptr,            \
                     T, NB, Hp, Wp, Ci, (int)Ho, (int)Wo, (int)Co)
  if (sbuf) LAUNCH_MMV2(false, V2_LDS_BYTES_SBUF);
  else LAUNCH_MMV2(true, V2_LDS_BYTES);
#undef LAUNCH_MMV2
  return {y, sums};
}

// dy [T, NB, Ho, Wo, F] bf16 ; x [T, NB, H, W, C] bf16
// -> {dw [T, F, C, 3, 3] fp32, db [T, F] fp32}   (db fused — the dY tile
// is already staged for the GEMM, so the bias reduction is nearly free)
std::vector<torch::Tensor> tconv_wgrad(torch::Tensor dy, torch::Tensor x,
                                       long pad, bool with_bias) {
  TORCH_CHECK(dy.is_cuda() && x.is_cuda());
  auto dyc = dy.contiguous();
  auto xc = x.contiguous();
  const int T = (int)x.size(0), NB = (int)x.size(1), H = (int)x.size(2),
            W = (int)x.size(3), C = (int)x.size(4);
  const int Ho = (int)dy.size(2), Wo = (int)dy.size(3), F = (int)dy.size(4);
  TORCH_CHECK(F <= 64, "F must be <= 64");
  const int N9 = 9 * C;
  const long Ktot = (long)NB * Ho * Wo;
  const char* det_env = getenv("MAML355_DETERMINISTIC");
  const bool det = det_env && det_env[0] == '1';
  // size K-chunks so the grid has >= ~1024 blocks (256 CUs want far more
  // workgroups than CUs; small support-pass K was leaving the chip idle)
  const int gridx = (N9 + WGN - 1) / WGN;
  long desired_y = std::max<long>(1, 1024 / std::max<long>(1, (long)gridx * T));
  long kchunk = (Ktot + desired_y - 1) / desired_y;
  kchunk = ((kchunk + WBK - 1) / WBK) * WBK;
  kchunk = std::max<long>(kchunk, WBK);
  if (!det) kchunk = std::min<long>(kchunk, WG_KCHUNK);
  const int gridy = (int)((Ktot + kchunk - 1) / kchunk);
  // MAML355_DETERMINISTIC: each (t, K-chunk) block writes a private
  // accumulator slice (no fp32 atomics); slices are summed in fixed order
  // by the finalize kernel -> bitwise run-to-run reproducible wgrad.
  const int nslices = det ? gridy : 1;
  auto acc = torch::zeros({T, nslices, N9, F},
                          x.options().dtype(torch::kFloat32));
  auto dbacc = torch::zeros({T, nslices, F},
                            x.options().dtype(torch::kFloat32));
  dim3 grid((unsigned)gridx, (unsigned)gridy, T);
  auto stream = at::cuda::getCurrentCUDAStream();
#define LAUNCH_WGRAD(DET_)                                                     \
  hipLaunchKernelGGL((tconv_wgrad_kernel<DET_>), grid, dim3(256), 0,           \
                     stream.stream(),                                          \
                     reinterpret_cast<const bf16*>(dyc.data_ptr()),            \
                     reinterpret_cast<const bf16*>(xc.data_ptr()),             \
                     acc.data_ptr<float>(),                                    \
                     with_bias ? dbacc.data_ptr<float>() : nullptr,            \
                     T, NB, H, W, C, Ho, Wo, F, (int)pad, (int)kchunk)
  if (det) LAUNCH_WGRAD(true); else LAUNCH_WGRAD(false);
#undef LAUNCH_WGRAD
  auto dw = torch::empty({T, F, C, 3, 3}, x.options().dtype(torch::kFloat32));
  const long total = (long)T * F * C * 9;
  hipLaunchKernelGGL(wgrad_finalize_kernel, dim3(ew_grid2(total, 256)),
                     dim3(256), 0, stream.stream(), acc.data_ptr<float>(),
                     dw.data_ptr<float>(), T, F, C, nslices);
  auto db = nslices == 1 ? dbacc.select(1, 0).contiguous() : dbacc.sum(1);
  return {dw, db};
}

// wgrad v2: transpose both operands into k-contiguous layouts, then the
// async-staged main kernel.  dy [T,NB,Ho,Wo,F], x [T,NB,H,W,C] ->
// {dw [T,F,C,3,3] fp32, db [T,F] fp32}.  Requires Wo <= 64, F <= 64.
std::vector<torch::Tensor> tconv_wgrad_v2(torch::Tensor dy, torch::Tensor x,
                                          long pad, bool with_bias) {
  TORCH_CHECK(dy.is_cuda() && x.is_cuda());
  auto dyc = dy.contiguous();
  auto xc = x.contiguous();
  const int T = (int)x.size(0), NB = (int)x.size(1), H = (int)x.size(2),
            W = (int)x.size(3), C = (int)x.size(4);
  const int Ho = (int)dy.size(2), Wo = (int)dy.size(3), F = (int)dy.size(4);
  TORCH_CHECK(F <= 64, "wgrad_v2 needs F <= 64");
  const char* det_env = getenv("MAML355_DETERMINISTIC");
  const bool det = det_env && det_env[0] == '1';
  const int N9 = 9 * C;
  const int Wo8 = ((Wo + 7) / 8) * 8;
  const long Kprime = (long)NB * Ho * Wo8;
  const long Kr = ((Kprime + WBK - 1) / WBK) * WBK;
  const int Hxp = H + 2 * (int)pad;
  const int Wxp = Wo8 + 2;
  auto stream = at::cuda::getCurrentCUDAStream();

  // operand transposes
  auto dyt = torch::empty({T, F, Kr}, x.options());
  {
    const long blocks = (long)T * NB * Ho;
    hipLaunchKernelGGL(dyt_pad_kernel, dim3((unsigned)blocks), dim3(256), 0,
                       stream.stream(),
                       reinterpret_cast<const bf16*>(dyc.data_ptr()),
                       reinterpret_cast<bf16*>(dyt.data_ptr()),
                       T, NB, Ho, Wo, F, Wo8, Kr);
    if (Kr > Kprime) {
      const long total = (long)T * F * (Kr - Kprime);
      hipLaunchKernelGGL(dyt_tail_kernel, dim3(ew_grid2(total, 256)), dim3(256),
                         0, stream.stream(),
                         reinterpret_cast<bf16*>(dyt.data_ptr()), T, F, Kprime,
                         Kr);
    }
  }
  auto xt = torch::empty({(long)T * C * NB * Hxp * Wxp}, x.options());
  {
    const long blocks = (long)T * NB * Hxp * ((Wxp + 63) / 64);
    hipLaunchKernelGGL(xt_pad_kernel, dim3((unsigned)blocks), dim3(256), 0,
                       stream.stream(),
                       reinterpret_cast<const bf16*>(xc.data_ptr()),
                       reinterpret_cast<bf16*>(xt.data_ptr()),
                       T, NB, H, W, C, Hxp, Wxp, (int)pad);
  }
  auto zpage = torch::zeros({16}, x.options());

  // n-columns per block: NSUB=2 (128 cols, 2x MFMA per barrier) pays only
  // on large-K shapes (measured: conv1-tgt 94->104 TF; SLOWER below
  // Kr ~1.3e5 — grid starvation + 48 KB LDS occupancy drop).  Env
  // MAML355_WGRAD_NSUB overrides for A/B.
  int nsub = (N9 > 64 && Kr >= 131072) ? 2 : 1;
  if (const char* e = getenv("MAML355_WGRAD_NSUB")) {
    if (e[0] == '1') nsub = 1; else if (e[0] == '2' && N9 > 64) nsub = 2;
  }
  const int wgn = 64 * nsub;

  // grid sizing as v1 (>= ~1024 blocks), K-chunks multiple of WBK
  const int gridx = (N9 + wgn - 1) / wgn;
  long desired_y = std::max<long>(1, 1024 / std::max<long>(1, (long)gridx * T));
  long kchunk = (Kr + desired_y - 1) / desired_y;
  kchunk = ((kchunk + WBK - 1) / WBK) * WBK;
  kchunk = std::max<long>(kchunk, WBK);
  if (!det) kchunk = std::min<long>(kchunk, (long)WG_KCHUNK);
  const int gridy = (int)((Kr + kchunk - 1) / kchunk);
  const int nslices = det ? gridy : 1;
  auto acc = torch::zeros({T, nslices, N9, F},
                          x.options().dtype(torch::kFloat32));
  auto dbacc = torch::zeros({T, nslices, F},
                            x.options().dtype(torch::kFloat32));
  dim3 grid((unsigned)gridx, (unsigned)gridy, T);
#define LAUNCH_WG2(NS_, DET_)                                                  \
  hipLaunchKernelGGL((tconv_wgrad_v2_kernel<NS_, DET_>), grid, dim3(256), 0,   \
                     stream.stream(),                                          \
                     reinterpret_cast<const bf16*>(dyt.data_ptr()),            \
                     reinterpret_cast<const bf16*>(xt.data_ptr()),             \
                     reinterpret_cast<const bf16*>(zpage.data_ptr()),          \
                     acc.data_ptr<float>(),                                    \
                     with_bias ? dbacc.data_ptr<float>() : nullptr,            \
                     T, NB, Ho, Wo8, Hxp, Wxp, C, F, Kr, Kprime, (int)kchunk)
  if (nsub == 2) {
    if (det) LAUNCH_WG2(2, true); else LAUNCH_WG2(2, false);
  } else {
    if (det) LAUNCH_WG2(1, true); else LAUNCH_WG2(1, false);
  }
#undef LAUNCH_WG2
  auto dw = torch::empty({T, F, C, 3, 3}, x.options().dtype(torch::kFloat32));
  const long total = (long)T * F * C * 9;
  hipLaunchKernelGGL(wgrad_finalize_kernel, dim3(ew_grid2(total, 256)),
                     dim3(256), 0, stream.stream(), acc.data_ptr<float>(),
                     dw.data_ptr<float>(), T, F, C, nslices);
  auto db = nslices == 1 ? dbacc.select(1, 0).contiguous() : dbacc.sum(1);
  return {dw, db};
}

std::vector<torch::Tensor> mfma_probe(torch::Tensor A, torch::Tensor B) {
  TORCH_CHECK(A.is_cuda() && A.sizes() == torch::IntArrayRef({16, 32}));
  TORCH_CHECK(B.sizes() == torch::IntArrayRef({32, 16}));
  auto Ac = A.contiguous().to(torch::kBFloat16);
  auto Bc = B.contiguous().to(torch::kBFloat16);
  auto D = torch::zeros({16, 16}, A.options().dtype(torch::kFloat32));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream.stream(),
                     reinterpret_cast<const bf16*>(Ac.data_ptr()),
                     reinterpret_cast<const bf16*>(Bc.data_ptr()),
                     D.data_ptr<float>());
  return {D};
}
