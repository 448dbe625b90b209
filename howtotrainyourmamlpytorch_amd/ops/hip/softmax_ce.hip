// Fused softmax + cross-entropy for few-shot logits [T, M, ways]
// (reference: F.cross_entropy at few_shot_learning_system.py:284).
// Forward: one wave per row -> per-task mean loss [T] in one pass
// (atomicAdd of row loss / M into loss[t]).  Saves softmax probs for the
// first-order backward kernel; the create_graph path recomputes in torch.

#include "common.h"
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

using namespace maml355;

__global__ void ce_fwd_kernel(const float* __restrict__ logits,
                              const long* __restrict__ labels,
                              float* __restrict__ probs,
                              float* __restrict__ loss,  // [T], pre-zeroed
                              int T, int M, int ways) {
  const long rows = (long)T * M;
  const int lane = threadIdx.x & (WAVE - 1);
  const long wave_id = ((long)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const long nwaves = grid_stride() / WAVE;
  for (long row = wave_id; row < rows; row += nwaves) {
    const float* lrow = logits + row * ways;
    float v = (lane < ways) ? lrow[lane] : -INFINITY;
    const float mx = wave_reduce_max(v);
    const float mxb = __shfl(mx, 0, WAVE);
    float e = (lane < ways) ? __expf(v - mxb) : 0.f;
    const float se = wave_reduce_sum(e);
    const float seb = __shfl(se, 0, WAVE);
    if (lane < ways) probs[row * ways + lane] = e / seb;
    if (lane == 0) {
      const long t = row / M;
      const long y = labels[row];
      const float row_loss = logf(seb) + mxb - lrow[y];
      atomicAdd(&loss[t], row_loss / (float)M);
    }
  }
}

__global__ void ce_bwd_kernel(const float* __restrict__ probs,
                              const long* __restrict__ labels,
                              const float* __restrict__ gtask,  // [T]
                              float* __restrict__ dlogits,
                              int T, int M, int ways) {
  const long total = (long)T * M * ways;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += grid_stride()) {
    const int w = (int)(i % ways);
    const long row = i / ways;
    const long t = row / M;
    const float onehot = (labels[row] == w) ? 1.f : 0.f;
    dlogits[i] = (probs[i] - onehot) * gtask[t] / (float)M;
  }
}

// Double-backward: dlogits = (p - onehot) * gt/M with p = softmax(logits).
// Given gdl = dL/d(dlogits):
//   d_logits = gt/M * (p .* gdl - p * <p, gdl>_row)     (softmax Jacobian)
//   d_gtask[t] = sum_rows <gdl, p - onehot> / M
__global__ void ce_dbwd_kernel(const float* __restrict__ probs,
                               const long* __restrict__ labels,
                               const float* __restrict__ gdl,
                               const float* __restrict__ gtask,
                               float* __restrict__ d_logits,
                               float* __restrict__ d_gtask,  // pre-zeroed [T]
                               int T, int M, int ways) {
  const long rows = (long)T * M;
  const int lane = threadIdx.x & (WAVE - 1);
  const long wave_id = ((long)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const long nwaves = grid_stride() / WAVE;
  for (long row = wave_id; row < rows; row += nwaves) {
    const long t = row / M;
    const float gt_over_M = gtask[t] / (float)M;
    const float p = (lane < ways) ? probs[row * ways + lane] : 0.f;
    const float g = (lane < ways) ? gdl[row * ways + lane] : 0.f;
    const float dot = wave_reduce_sum(p * g);
    const float dotb = __shfl(dot, 0, WAVE);
    if (lane < ways) {
      d_logits[row * ways + lane] = gt_over_M * (p * g - p * dotb);
    }
    if (lane == 0) {
      const float gl = gdl[row * ways + labels[row]];
      atomicAdd(&d_gtask[t], (dotb - gl) / (float)M);
    }
  }
}

static int grid_for(long total, int threads) {
  long blocks = (total + threads - 1) / threads;
  return (int)std::min<long>(blocks, 4096);
}

std::vector<torch::Tensor> ce_fwd(torch::Tensor logits, torch::Tensor labels) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 3);
  TORCH_CHECK(logits.size(2) <= WAVE, "ways must be <= 64");
  auto lf = logits.contiguous().to(torch::kFloat32);
  auto lab = labels.contiguous().to(torch::kLong);
  const int T = (int)logits.size(0), M = (int)logits.size(1),
            ways = (int)logits.size(2);
  auto probs = torch::empty_like(lf);
  auto loss = torch::zeros({T}, lf.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  const long rows = (long)T * M;
  hipLaunchKernelGGL(ce_fwd_kernel, dim3(grid_for(rows * WAVE, 256)), dim3(256),
                     0, stream.stream(), lf.data_ptr<float>(),
                     lab.data_ptr<long>(), probs.data_ptr<float>(),
                     loss.data_ptr<float>(), T, M, ways);
  return {loss, probs};
}

std::vector<torch::Tensor> ce_dbwd(torch::Tensor probs, torch::Tensor labels,
                                   torch::Tensor gdl, torch::Tensor gtask) {
  const int T = (int)probs.size(0), M = (int)probs.size(1),
            ways = (int)probs.size(2);
  auto d_logits = torch::empty_like(probs);
  auto d_gtask = torch::zeros({T}, probs.options());
  auto lab = labels.contiguous().to(torch::kLong);
  auto gdlc = gdl.contiguous().to(torch::kFloat32);
  auto gtc = gtask.contiguous().to(torch::kFloat32);
  auto stream = at::cuda::getCurrentCUDAStream();
  const long rows = (long)T * M;
  hipLaunchKernelGGL(ce_dbwd_kernel, dim3(grid_for(rows * WAVE, 256)),
                     dim3(256), 0, stream.stream(), probs.data_ptr<float>(),
                     lab.data_ptr<long>(), gdlc.data_ptr<float>(),
                     gtc.data_ptr<float>(), d_logits.data_ptr<float>(),
                     d_gtask.data_ptr<float>(), T, M, ways);
  return {d_logits, d_gtask};
}

torch::Tensor ce_bwd(torch::Tensor probs, torch::Tensor labels,
                     torch::Tensor gtask) {
  const int T = (int)probs.size(0), M = (int)probs.size(1),
            ways = (int)probs.size(2);
  auto dlogits = torch::empty_like(probs);
  auto gt = gtask.contiguous().to(torch::kFloat32);
  auto lab = labels.contiguous().to(torch::kLong);
  auto stream = at::cuda::getCurrentCUDAStream();
  const long total = (long)T * M * ways;
  hipLaunchKernelGGL(ce_bwd_kernel, dim3(grid_for(total, 256)), dim3(256), 0,
                     stream.stream(), probs.data_ptr<float>(),
                     lab.data_ptr<long>(), gt.data_ptr<float>(),
                     dlogits.data_ptr<float>(), T, M, ways);
  return dlogits;
}
