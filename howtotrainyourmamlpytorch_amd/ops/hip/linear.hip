// Task-batched linear head (reference: F.linear via
// meta_neural_network_architectures.py:141) — the LAST library op on the
// training path.  torch.bmm routes to hipBLASLt, whose first-call
// autotuning made run #1 of a process numerically different from runs
// #2+ (measured: identical 5e-3 theta drift after 3 Adam steps, gone on
// CPU).  These kernels are pure functions of their inputs: deterministic
// across calls, processes and allocator states.
//
// Math (per task):  y = x @ w^T + b     x [M,K] bf16, w [ways,K] fp32
//   dx = dy @ w        dw = dy^T @ x        db = sum_m dy
// The three ops are mutually bilinear, so each backward composes the
// other two — custom kernels at every derivative order (second-order
// MAML included), mirroring the conv trio.
// w/b are read in fp32 and ROUNDED to bf16 in-kernel — numerically
// identical to the previous bmm-on-bf16-cast path, with fp32 dw/db out.
// Shapes are tiny (M <= ~500, ways <= 64, K <= ~3200): VALU kernels.

#include "common.h"
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

using namespace maml355;

using bf16 = __hip_bfloat16;

DEVINL float rb(float v) {  // round fp32 -> bf16 -> fp32
  return __bfloat162float(__float2bfloat16(v));
}
DEVINL float ld_bf(const short* p) {
  return __bfloat162float(__hip_bfloat16(__hip_bfloat16_raw{(unsigned short)*p}));
}

// y[t,m,w] = sum_k x[t,m,k] * bf16(w[t,w,k]) + bf16(b[t,w]); one wave per
// (t, m) row, lanes k-parallel with a wave reduction per way.
__global__ void lin_fwd_kernel(const bf16* __restrict__ X,
                               const float* __restrict__ W,
                               const float* __restrict__ B,
                               bf16* __restrict__ Y,
                               int T, int M, int K, int ways) {
  const int lane = threadIdx.x & (WAVE - 1);
  const long rows = (long)T * M;
  const long wave_id = ((long)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const long nwaves = grid_stride() / WAVE;
  for (long row = wave_id; row < rows; row += nwaves) {
    const long t = row / M;
    const short* xr = (const short*)X + row * K;
    for (int w = 0; w < ways; ++w) {
      const float* wr = W + (t * ways + w) * (long)K;
      float acc = 0.f;
      for (int k = lane; k < K; k += WAVE) {
        acc = fmaf(ld_bf(xr + k), rb(wr[k]), acc);
      }
      acc = wave_reduce_sum(acc);
      if (lane == 0) {
        if (B) acc += rb(B[t * ways + w]);
        ((short*)Y)[row * ways + w] =
            (short)__bfloat16_as_short(__float2bfloat16(acc));
      }
    }
  }
}

// dx[t,m,k] = sum_w dy[t,m,w] * bf16(W[t,w,k]); thread per element.
__global__ void lin_dx_kernel(const bf16* __restrict__ dY,
                              const float* __restrict__ W,
                              bf16* __restrict__ dX,
                              int T, int M, int K, int ways) {
  const long total = (long)T * M * K;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += grid_stride()) {
    const int k = (int)(i % K);
    const long row = i / K;
    const long t = row / M;
    const short* dyr = (const short*)dY + row * ways;
    float acc = 0.f;
    for (int w = 0; w < ways; ++w) {
      acc = fmaf(ld_bf(dyr + w), rb(W[(t * ways + w) * (long)K + k]), acc);
    }
    ((short*)dX)[i] = (short)__bfloat16_as_short(__float2bfloat16(acc));
  }
}

// dw[t,w,k] = sum_m dy[t,m,w] * x[t,m,k] (fp32 out, fixed m order ->
// deterministic); thread per element with a serial m loop.
__global__ void lin_wgrad_kernel(const bf16* __restrict__ dY,
                                 const bf16* __restrict__ X,
                                 float* __restrict__ dW,
                                 int T, int M, int K, int ways) {
  const long total = (long)T * ways * K;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += grid_stride()) {
    const int k = (int)(i % K);
    const int w = (int)((i / K) % ways);
    const long t = i / ((long)K * ways);
    const short* dyt = (const short*)dY + t * (long)M * ways + w;
    const short* xt = (const short*)X + t * (long)M * K + k;
    float acc = 0.f;
    for (int m = 0; m < M; ++m) {
      acc = fmaf(ld_bf(dyt + (long)m * ways), ld_bf(xt + (long)m * K), acc);
    }
    dW[i] = acc;
  }
}

// db[t,w] = sum_m dy[t,m,w] (fp32, fixed order)
__global__ void lin_db_kernel(const bf16* __restrict__ dY,
                              float* __restrict__ dB,
                              int T, int M, int ways) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= T * ways) return;
  const int w = i % ways;
  const long t = i / ways;
  const short* dyt = (const short*)dY + t * (long)M * ways + w;
  float acc = 0.f;
  for (int m = 0; m < M; ++m) acc += ld_bf(dyt + (long)m * ways);
  dB[i] = acc;
}

static int lgrid(long total, int threads) {
  long b = (total + threads - 1) / threads;
  return (int)std::min<long>(b, 4096);
}

torch::Tensor lin_fwd(torch::Tensor x, torch::Tensor w,
                      c10::optional<torch::Tensor> b) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 3 && x.scalar_type() == torch::kBFloat16);
  auto xc = x.contiguous();
  auto wc = w.contiguous().to(torch::kFloat32);
  const int T = (int)x.size(0), M = (int)x.size(1), K = (int)x.size(2);
  const int ways = (int)w.size(1);
  TORCH_CHECK(w.size(0) == T && w.size(2) == K);
  auto y = torch::empty({T, M, ways}, x.options());
  const float* bptr = nullptr;
  torch::Tensor bc;
  if (b.has_value()) {
    bc = b->contiguous().to(torch::kFloat32);
    bptr = bc.data_ptr<float>();
  }
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(lin_fwd_kernel, dim3(lgrid((long)T * M * WAVE, 256)),
                     dim3(256), 0, stream.stream(),
                     reinterpret_cast<const bf16*>(xc.data_ptr()),
                     wc.data_ptr<float>(), bptr,
                     reinterpret_cast<bf16*>(y.data_ptr()), T, M, K, ways);
  return y;
}

torch::Tensor lin_dx(torch::Tensor dy, torch::Tensor w) {
  auto dyc = dy.contiguous();
  auto wc = w.contiguous().to(torch::kFloat32);
  const int T = (int)dy.size(0), M = (int)dy.size(1), ways = (int)dy.size(2);
  const int K = (int)w.size(2);
  auto dx = torch::empty({T, M, K}, dy.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(lin_dx_kernel, dim3(lgrid((long)T * M * K, 256)),
                     dim3(256), 0, stream.stream(),
                     reinterpret_cast<const bf16*>(dyc.data_ptr()),
                     wc.data_ptr<float>(),
                     reinterpret_cast<bf16*>(dx.data_ptr()), T, M, K, ways);
  return dx;
}

std::vector<torch::Tensor> lin_wgrad(torch::Tensor dy, torch::Tensor x,
                                     bool with_bias) {
  auto dyc = dy.contiguous();
  auto xc = x.contiguous();
  const int T = (int)dy.size(0), M = (int)dy.size(1), ways = (int)dy.size(2);
  const int K = (int)x.size(2);
  auto dw = torch::empty({T, ways, K}, x.options().dtype(torch::kFloat32));
  auto db = torch::empty({T, ways}, x.options().dtype(torch::kFloat32));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(lin_wgrad_kernel, dim3(lgrid((long)T * ways * K, 256)),
                     dim3(256), 0, stream.stream(),
                     reinterpret_cast<const bf16*>(dyc.data_ptr()),
                     reinterpret_cast<const bf16*>(xc.data_ptr()),
                     dw.data_ptr<float>(), T, M, K, ways);
  if (with_bias) {
    hipLaunchKernelGGL(lin_db_kernel, dim3((T * ways + 255) / 256), dim3(256),
                       0, stream.stream(),
                       reinterpret_cast<const bf16*>(dyc.data_ptr()),
                       db.data_ptr<float>(), T, M, ways);
  }
  return {dw, db};
}
