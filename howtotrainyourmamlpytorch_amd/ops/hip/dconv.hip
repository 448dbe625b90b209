// Direct (non-GEMM) 3x3 convolution for SMALL input-channel counts — the
// network's first layer (C in {1,3}; reference images are grayscale
// Omniglot / RGB mini-imagenet).
//
// The im2col-MFMA path is mis-shaped here: K = 9*C is 9..27, so the
// 32-wide MFMA K granularity plus BK=64 staging wastes >4x the real MACs
// and the measured rate is 8-25 TF.  CDNA4's VECTOR ALU (157 TF fp32) is
// the right unit for K this small:
//
//   fwd:   one thread per output position computes ALL F channels with
//          the 9*C*F weights staged in LDS and read wave-uniform
//          (hardware broadcast — no bank traffic), x window in registers.
//   wgrad: lane = output channel f; each wave walks a contiguous range of
//          positions; dy reads are coalesced across lanes, the x window
//          is a wave-uniform (single-cacheline) load; each lane holds its
//          f's full dW[c][ky][kx] tile (<= 27 fp32) in registers.
//
// Both are FMA-throughput bound by construction instead of staging bound.

#include "common.h"
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

using namespace maml355;

using bf16 = __hip_bfloat16;
typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8;

// ---------------------------------------------------------------------------
// fwd: X [T,NB,H,W,C] bf16 -> Y [T,NB,Ho,Wo,F] bf16 (F = FT compile-time)
// ---------------------------------------------------------------------------
template <int FT>
__global__ __launch_bounds__(256)
void dconv_fwd_kernel(const bf16* __restrict__ X, const float* __restrict__ W,
                      const float* __restrict__ bias, bf16* __restrict__ Y,
                      int T, int NB, int H, int Wd, int C,
                      int Ho, int Wo, int pad) {
  const int t = blockIdx.y;
  __shared__ float wl[9 * 8 * FT];  // [kyx][c][f]
  const float* Wt = W + (long)t * FT * C * 9;
  for (int i = threadIdx.x; i < FT * C * 9; i += blockDim.x) {
    // i over [f][c][kyx] source order -> [kyx][c][f] LDS order
    const int kyx = i % 9;
    const int c = (i / 9) % C;
    const int f = i / (9 * C);
    wl[(kyx * C + c) * FT + f] = Wt[i];
  }
  __syncthreads();

  const long Mtot = (long)NB * Ho * Wo;
  const bf16* Xt = X + (long)t * NB * H * Wd * C;
  bf16* Yt = Y + (long)t * Mtot * FT;
  const float* bt = bias ? bias + (long)t * FT : nullptr;

  for (long m = (long)blockIdx.x * blockDim.x + threadIdx.x; m < Mtot;
       m += (long)blockDim.x * gridDim.x) {
    const int wo = (int)(m % Wo);
    const int ho = (int)((m / Wo) % Ho);
    const int n = (int)(m / ((long)Wo * Ho));
    float acc[FT];
#pragma unroll
    for (int f = 0; f < FT; ++f) acc[f] = bt ? bt[f] : 0.f;
    for (int ky = 0; ky < 3; ++ky) {
      const int h = ho + ky - pad;
      if (h < 0 || h >= H) continue;
      for (int kx = 0; kx < 3; ++kx) {
        const int w = wo + kx - pad;
        if (w < 0 || w >= Wd) continue;
        const short* xrow = (const short*)Xt + (((long)n * H + h) * Wd + w) * C;
        for (int c = 0; c < C; ++c) {
          const float xv = __bfloat162float(
              __hip_bfloat16(__hip_bfloat16_raw{(unsigned short)xrow[c]}));
          const float* wp = &wl[((ky * 3 + kx) * C + c) * FT];
#pragma unroll
          for (int f = 0; f < FT; ++f) acc[f] = fmaf(xv, wp[f], acc[f]);
        }
      }
    }
#pragma unroll
    for (int f8 = 0; f8 < FT / 8; ++f8) {
      bf16x8 v;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        v[j] = (short)__bfloat16_as_short(__float2bfloat16(acc[f8 * 8 + j]));
      *(bf16x8*)&((short*)Yt)[m * FT + f8 * 8] = v;
    }
  }
}

// ---------------------------------------------------------------------------
// wgrad: dWacc [T, nsl, 9C, F] (+ dBacc [T, nsl, F]) — same accumulator
// layout as the GEMM wgrad so the shared finalize kernel applies.
// lane = f; wave walks contiguous positions; CT = compile-time C.
// ---------------------------------------------------------------------------
template <int CT, bool DET>
__global__ __launch_bounds__(256)
void dconv_wgrad_kernel(const bf16* __restrict__ dY, const bf16* __restrict__ X,
                        float* __restrict__ dWacc, float* __restrict__ dBacc,
                        int T, int NB, int H, int Wd, int Ho, int Wo, int F,
                        int pad, long chunk) {
  const int t = blockIdx.y;
  const long Mtot = (long)NB * Ho * Wo;
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int nwaves = blockDim.x / WAVE;
  // this wave's contiguous position range
  const long c0 = blockIdx.x * chunk + wave * (chunk / nwaves);
  const long c1 = min(c0 + chunk / nwaves, Mtot);

  const bf16* Xt = X + (long)t * NB * H * Wd * CT;
  const short* dYt = (const short*)dY + (long)t * Mtot * F;

  float acc[9 * CT];
#pragma unroll
  for (int i = 0; i < 9 * CT; ++i) acc[i] = 0.f;
  float db = 0.f;

  if (c0 < Mtot) {
    int wo = (int)(c0 % Wo);
    int ho = (int)((c0 / Wo) % Ho);
    int n = (int)(c0 / ((long)Wo * Ho));
    for (long m = c0; m < c1; ++m) {
      const float dyv =
          (lane < F)
              ? __bfloat162float(__hip_bfloat16(
                    __hip_bfloat16_raw{(unsigned short)dYt[m * F + lane]}))
              : 0.f;
      db += dyv;
#pragma unroll
      for (int ky = 0; ky < 3; ++ky) {
        const int h = ho + ky - pad;
        if (h < 0 || h >= H) continue;
#pragma unroll
        for (int kx = 0; kx < 3; ++kx) {
          const int w = wo + kx - pad;
          if (w < 0 || w >= Wd) continue;
          const short* xrow =
              (const short*)Xt + (((long)n * H + h) * Wd + w) * CT;
#pragma unroll
          for (int c = 0; c < CT; ++c) {
            const float xv = __bfloat162float(__hip_bfloat16(
                __hip_bfloat16_raw{(unsigned short)xrow[c]}));
            acc[(ky * 3 + kx) * CT + c] = fmaf(dyv, xv, acc[(ky * 3 + kx) * CT + c]);
          }
        }
      }
      // incremental position decode
      if (++wo == Wo) {
        wo = 0;
        if (++ho == Ho) { ho = 0; ++n; }
      }
    }
  }

  // ordered in-block reduction across waves via LDS, then one
  // store (DET slice) or atomicAdd per [9C][F] element per block
  float* dWt = DET ? dWacc + ((long)t * gridDim.x + blockIdx.x) * 9 * CT * F
                   : dWacc + (long)t * 9 * CT * F;
  {
    __shared__ float red[256];
#pragma unroll
    for (int i = 0; i < 9 * CT; ++i) {
      red[threadIdx.x] = acc[i];
      __syncthreads();
      if (wave == 0 && lane < F) {
        float s = 0.f;
        for (int wv = 0; wv < nwaves; ++wv) s += red[wv * WAVE + lane];
        if (DET) {
          dWt[(long)i * F + lane] = s;
        } else {
          atomicAdd(&dWt[(long)i * F + lane], s);
        }
      }
      __syncthreads();
    }
  }
  if (dBacc != nullptr) {
    __shared__ float dbl[256];
    dbl[threadIdx.x] = db;
    __syncthreads();
    if (wave == 0 && lane < F) {
      float s = 0.f;
      for (int wv = 0; wv < nwaves; ++wv) s += dbl[wv * WAVE + lane];
      if (DET) {
        dBacc[((long)t * gridDim.x + blockIdx.x) * F + lane] = s;
      } else {
        atomicAdd(&dBacc[(long)t * F + lane], s);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------
// local copy of the slice-summing finalize (same as tconv.hip's — kept
// per-TU because the build is non-RDC)
__global__ void dconv_finalize_kernel(const float* __restrict__ acc,
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
    for (int s = 0; s < nslices; ++s) v += acc[(t * nslices + s) * ssz + e];
    dw[i] = v;
  }
}

torch::Tensor dconv_fwd(torch::Tensor x, torch::Tensor w,
                        c10::optional<torch::Tensor> bias, long pad,
                        long Ho, long Wo) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 5 && x.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  const int T = (int)x.size(0), NB = (int)x.size(1), H = (int)x.size(2),
            W = (int)x.size(3), C = (int)x.size(4);
  const int F = (int)w.size(1);
  TORCH_CHECK(C <= 8 && (F == 16 || F == 32 || F == 48 || F == 64),
              "dconv_fwd needs C<=8 and F in {16,32,48,64}");
  auto wc = w.contiguous().to(torch::kFloat32);
  auto y = torch::empty({T, NB, Ho, Wo, F}, x.options());
  const float* bptr = nullptr;
  torch::Tensor bc;
  if (bias.has_value()) {
    bc = bias->contiguous().to(torch::kFloat32);
    bptr = bc.data_ptr<float>();
  }
  const long Mtot = (long)NB * Ho * Wo;
  const int blocks = (int)std::min<long>((Mtot + 255) / 256, 4096);
  auto stream = at::cuda::getCurrentCUDAStream();
#define LAUNCH_DFWD(FT)                                                        \
  hipLaunchKernelGGL((dconv_fwd_kernel<FT>), dim3(blocks, T), dim3(256), 0,    \
                     stream.stream(),                                          \
                     reinterpret_cast<const bf16*>(x.data_ptr()),              \
                     wc.data_ptr<float>(), bptr,                               \
                     reinterpret_cast<bf16*>(y.data_ptr()),                    \
                     T, NB, H, W, C, (int)Ho, (int)Wo, (int)pad)
  switch (F) {
    case 16: LAUNCH_DFWD(16); break;
    case 32: LAUNCH_DFWD(32); break;
    case 48: LAUNCH_DFWD(48); break;
    default: LAUNCH_DFWD(64); break;
  }
#undef LAUNCH_DFWD
  return y;
}

std::vector<torch::Tensor> dconv_wgrad(torch::Tensor dy, torch::Tensor x,
                                       long pad, bool with_bias) {
  TORCH_CHECK(dy.is_cuda() && x.is_cuda());
  auto dyc = dy.contiguous();
  auto xc = x.contiguous();
  const int T = (int)x.size(0), NB = (int)x.size(1), H = (int)x.size(2),
            W = (int)x.size(3), C = (int)x.size(4);
  const int Ho = (int)dy.size(2), Wo = (int)dy.size(3), F = (int)dy.size(4);
  TORCH_CHECK(C <= 8 && F <= 64, "dconv_wgrad needs C<=8, F<=64");
  const char* det_env = getenv("MAML355_DETERMINISTIC");
  const bool det = det_env && det_env[0] == '1';
  const long Mtot = (long)NB * Ho * Wo;
  // chunk so total waves fill the chip (4 waves per block)
  long gridx = std::max<long>(1, 2048 / std::max(1, T));
  long chunk = (Mtot + gridx - 1) / gridx;
  chunk = std::max<long>(((chunk + 3) / 4) * 4, 4);
  gridx = (Mtot + chunk - 1) / chunk;
  const int nslices = det ? (int)gridx : 1;
  auto acc = torch::zeros({T, nslices, 9 * C, F},
                          x.options().dtype(torch::kFloat32));
  auto dbacc = torch::zeros({T, nslices, F},
                            x.options().dtype(torch::kFloat32));
  auto stream = at::cuda::getCurrentCUDAStream();
#define LAUNCH_DWG(CT, DET_)                                                   \
  hipLaunchKernelGGL((dconv_wgrad_kernel<CT, DET_>), dim3((unsigned)gridx, T), \
                     dim3(256), 0, stream.stream(),                            \
                     reinterpret_cast<const bf16*>(dyc.data_ptr()),            \
                     reinterpret_cast<const bf16*>(xc.data_ptr()),             \
                     acc.data_ptr<float>(),                                    \
                     with_bias ? dbacc.data_ptr<float>() : nullptr,            \
                     T, NB, H, W, Ho, Wo, F, (int)pad, chunk)
#define LAUNCH_DWG_DET(CT)                                                     \
  do { if (det) LAUNCH_DWG(CT, true); else LAUNCH_DWG(CT, false); } while (0)
  switch (C) {
    case 1: LAUNCH_DWG_DET(1); break;
    case 2: LAUNCH_DWG_DET(2); break;
    case 3: LAUNCH_DWG_DET(3); break;
    case 4: LAUNCH_DWG_DET(4); break;
    default: LAUNCH_DWG_DET(8); break;
  }
#undef LAUNCH_DWG
#undef LAUNCH_DWG_DET
  auto dw = torch::empty({T, F, C, 3, 3}, x.options().dtype(torch::kFloat32));
  const long total = (long)T * F * C * 9;
  const int fb = (int)std::min<long>((total + 255) / 256, 4096);
  hipLaunchKernelGGL(dconv_finalize_kernel, dim3(fb), dim3(256), 0,
                     stream.stream(), acc.data_ptr<float>(),
                     dw.data_ptr<float>(), T, F, C, nslices);
  auto db = nslices == 1 ? dbacc.select(1, 0).contiguous() : dbacc.sum(1);
  return {dw, db};
}
