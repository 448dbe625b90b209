// Task-batched NHWC 2x2/stride-2 max pooling (floor mode), forward +
// backward.  Replaces the reference's F.max_pool2d
// (meta_neural_network_architectures.py:605,651-652); the backward scatters
// via a saved 2-bit argmax mask with no atomics (windows are disjoint).

#include "common.h"
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

using namespace maml355;

template <typename scalar_t>
__global__ void maxpool_fwd_kernel(const scalar_t* __restrict__ x,
                                   scalar_t* __restrict__ y,
                                   unsigned char* __restrict__ mask,
                                   long N, int H, int W, int C, int Ho, int Wo) {
  const long total = N * (long)Ho * Wo * C;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += grid_stride()) {
    const int c = (int)(i % C);
    long r = i / C;
    const int wo = (int)(r % Wo); r /= Wo;
    const int ho = (int)(r % Ho); r /= Ho;
    const long n = r;
    const long base = ((n * H + 2 * ho) * W + 2 * wo) * C + c;
    float best = to_f32(x[base]);
    int arg = 0;
    const float v01 = to_f32(x[base + C]);
    if (v01 > best) { best = v01; arg = 1; }
    const float v10 = to_f32(x[base + (long)W * C]);
    if (v10 > best) { best = v10; arg = 2; }
    const float v11 = to_f32(x[base + (long)W * C + C]);
    if (v11 > best) { best = v11; arg = 3; }
    y[i] = from_f32<scalar_t>(best);
    mask[i] = (unsigned char)arg;
  }
}

// gather-style backward: one thread per INPUT element; positions outside the
// pooled region (odd H/W tails) get 0 — no memset, fully coalesced writes.
template <typename scalar_t>
__global__ void maxpool_bwd_kernel(const scalar_t* __restrict__ dy,
                                   const unsigned char* __restrict__ mask,
                                   scalar_t* __restrict__ dx,
                                   long N, int H, int W, int C, int Ho, int Wo) {
  const long total = N * (long)H * W * C;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += grid_stride()) {
    const int c = (int)(i % C);
    long r = i / C;
    const int w = (int)(r % W); r /= W;
    const int h = (int)(r % H); r /= H;
    const long n = r;
    const int ho = h >> 1, wo = w >> 1;
    float g = 0.f;
    if (ho < Ho && wo < Wo) {
      const long o = ((n * Ho + ho) * Wo + wo) * C + c;
      const int arg = ((h & 1) << 1) | (w & 1);
      if ((int)mask[o] == arg) g = to_f32(dy[o]);
    }
    dx[i] = from_f32<scalar_t>(g);
  }
}

// --------------------------------------------------------------------------
// vectorized variants (C % 8 == 0): one thread per (position, 8-channel
// group) — bf16x8 loads/stores, 8-byte mask packs.
// --------------------------------------------------------------------------
typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8p;

template <typename scalar_t>
__global__ void maxpool_fwd_vec_kernel(const scalar_t* __restrict__ x,
                                       scalar_t* __restrict__ y,
                                       unsigned char* __restrict__ mask,
                                       long N, int H, int W, int C,
                                       int Ho, int Wo) {
  const int c8n = C / 8;
  const long total = N * (long)Ho * Wo * c8n;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += grid_stride()) {
    const int c8 = (int)(i % c8n);
    long r = i / c8n;
    const int wo = (int)(r % Wo); r /= Wo;
    const int ho = (int)(r % Ho); r /= Ho;
    const long n = r;
    const long base = ((n * H + 2 * ho) * W + 2 * wo) * C + c8 * 8;
    float v00[8], v01[8], v10[8], v11[8];
    load8(&x[base], v00);
    load8(&x[base + C], v01);
    load8(&x[base + (long)W * C], v10);
    load8(&x[base + (long)W * C + C], v11);
    const long o = ((n * Ho + ho) * Wo + wo) * C + c8 * 8;
    unsigned long long mpack = 0;
    float out[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float best = v00[j];
      int arg = 0;
      if (v01[j] > best) { best = v01[j]; arg = 1; }
      if (v10[j] > best) { best = v10[j]; arg = 2; }
      if (v11[j] > best) { best = v11[j]; arg = 3; }
      out[j] = best;
      mpack |= ((unsigned long long)arg) << (8 * j);
    }
    store8(&y[o], out);
    *(unsigned long long*)&mask[o] = mpack;
  }
}

template <typename scalar_t>
__global__ void maxpool_bwd_vec_kernel(const scalar_t* __restrict__ dy,
                                       const unsigned char* __restrict__ mask,
                                       scalar_t* __restrict__ dx,
                                       long N, int H, int W, int C,
                                       int Ho, int Wo) {
  const int c8n = C / 8;
  const long total = N * (long)H * W * c8n;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += grid_stride()) {
    const int c8 = (int)(i % c8n);
    long r = i / c8n;
    const int w = (int)(r % W); r /= W;
    const int h = (int)(r % H); r /= H;
    const long n = r;
    const int ho = h >> 1, wo = w >> 1;
    const long ibase = ((n * H + h) * W + w) * C + c8 * 8;
    if (ho < Ho && wo < Wo) {
      const long o = ((n * Ho + ho) * Wo + wo) * C + c8 * 8;
      const int arg = ((h & 1) << 1) | (w & 1);
      const unsigned long long mpack = *(const unsigned long long*)&mask[o];
      float dv[8], out[8];
      load8(&dy[o], dv);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int mj = (int)((mpack >> (8 * j)) & 0xff);
        out[j] = (mj == arg) ? dv[j] : 0.f;
      }
      store8(&dx[ibase], out);
    } else {
      float z[8] = {0, 0, 0, 0, 0, 0, 0, 0};
      store8(&dx[ibase], z);
    }
  }
}

static int ew_grid(long total, int threads) {
  long blocks = (total + threads - 1) / threads;
  return (int)std::min<long>(blocks, 4096);
}

// x: [N, H, W, C] contiguous (caller flattens T*NS -> N)
std::vector<torch::Tensor> maxpool2x2_fwd(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.is_contiguous());
  const long N = x.size(0);
  const int H = (int)x.size(1), W = (int)x.size(2), C = (int)x.size(3);
  const int Ho = H / 2, Wo = W / 2;
  auto y = torch::empty({N, Ho, Wo, C}, x.options());
  auto mask = torch::empty({N, Ho, Wo, C}, x.options().dtype(torch::kUInt8));
  const bool vec = (C % 8 == 0);
  const long total = N * (long)Ho * Wo * (vec ? C / 8 : C);
  auto stream = at::cuda::getCurrentCUDAStream();
#define LAUNCH_PF(st)                                                          \
  do {                                                                         \
    if (vec)                                                                   \
      hipLaunchKernelGGL((maxpool_fwd_vec_kernel<st>),                         \
                         dim3(ew_grid(total, 256)), dim3(256), 0,              \
                         stream.stream(),                                      \
                         reinterpret_cast<const st*>(x.data_ptr()),            \
                         reinterpret_cast<st*>(y.data_ptr()),                  \
                         mask.data_ptr<unsigned char>(), N, H, W, C, Ho, Wo);  \
    else                                                                       \
      hipLaunchKernelGGL((maxpool_fwd_kernel<st>),                             \
                         dim3(ew_grid(total, 256)), dim3(256), 0,              \
                         stream.stream(),                                      \
                         reinterpret_cast<const st*>(x.data_ptr()),            \
                         reinterpret_cast<st*>(y.data_ptr()),                  \
                         mask.data_ptr<unsigned char>(), N, H, W, C, Ho, Wo);  \
  } while (0)
  if (x.scalar_type() == torch::kFloat32) LAUNCH_PF(float);
  else if (x.scalar_type() == torch::kBFloat16) LAUNCH_PF(__hip_bfloat16);
  else TORCH_CHECK(false, "maxpool2x2_fwd: unsupported dtype");
#undef LAUNCH_PF
  return {y, mask};
}

torch::Tensor maxpool2x2_bwd(torch::Tensor dy, torch::Tensor mask,
                             long H, long W) {
  TORCH_CHECK(dy.is_cuda() && dy.dim() == 4);
  auto dyc = dy.contiguous();
  const long N = dy.size(0);
  const int Ho = (int)dy.size(1), Wo = (int)dy.size(2), C = (int)dy.size(3);
  auto dx = torch::empty({N, H, W, C}, dy.options());
  const bool vec = (C % 8 == 0);
  const long total = N * H * W * (vec ? C / 8 : C);
  auto stream = at::cuda::getCurrentCUDAStream();
#define LAUNCH_PB(st)                                                          \
  do {                                                                         \
    if (vec)                                                                   \
      hipLaunchKernelGGL((maxpool_bwd_vec_kernel<st>),                         \
                         dim3(ew_grid(total, 256)), dim3(256), 0,              \
                         stream.stream(),                                      \
                         reinterpret_cast<const st*>(dyc.data_ptr()),          \
                         mask.data_ptr<unsigned char>(),                       \
                         reinterpret_cast<st*>(dx.data_ptr()),                 \
                         N, (int)H, (int)W, C, Ho, Wo);                        \
    else                                                                       \
      hipLaunchKernelGGL((maxpool_bwd_kernel<st>),                             \
                         dim3(ew_grid(total, 256)), dim3(256), 0,              \
                         stream.stream(),                                      \
                         reinterpret_cast<const st*>(dyc.data_ptr()),          \
                         mask.data_ptr<unsigned char>(),                       \
                         reinterpret_cast<st*>(dx.data_ptr()),                 \
                         N, (int)H, (int)W, C, Ho, Wo);                        \
  } while (0)
  if (dy.scalar_type() == torch::kFloat32) LAUNCH_PB(float);
  else if (dy.scalar_type() == torch::kBFloat16) LAUNCH_PB(__hip_bfloat16);
  else TORCH_CHECK(false, "maxpool2x2_bwd: unsupported dtype");
#undef LAUNCH_PB
  return dx;
}
