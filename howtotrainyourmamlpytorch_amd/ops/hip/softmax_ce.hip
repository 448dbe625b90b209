// Fused softmax + cross-entropy for few-shot logits [T, M, ways]
// (reference: F.cross_entropy at few_shot_learning_system.py:284).
// Forward: one wave per row -> per-row losses, then an ordered per-task
// mean (deterministic; no fp32 atomics).  Saves softmax probs for the
// backward kernel; double-backward is the analytic softmax Jacobian.

#include "common.h"
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

using namespace maml355;

__global__ void ce_fwd_kernel(const float* __restrict__ logits,
                              const long* __restrict__ labels,
                              float* __restrict__ probs,
                              float* __restrict__ row_loss,  // [T*M]
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
      const long y = labels[row];
      row_loss[row] = logf(seb) + mxb - lrow[y];
    }
  }
}

// ordered per-task mean of row values [T, M] -> out [T] (deterministic —
// the fp32-atomic version made the task loss order-dependent)
__global__ void ce_rowmean_kernel(const float* __restrict__ rows,
                                  float* __restrict__ out, int T, int M) {
  const int t = blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= T) return;
  float s = 0.f;
  const float* r = rows + (long)t * M;
  for (int m = 0; m < M; ++m) s += r[m];
  out[t] = s / (float)M;
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
                               float* __restrict__ d_gtask_rows,  // [T*M]
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
      d_gtask_rows[row] = dotb - gl;
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
  auto loss = torch::empty({T}, lf.options());
  auto rowloss = torch::empty({T, M}, lf.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  const long rows = (long)T * M;
  hipLaunchKernelGGL(ce_fwd_kernel, dim3(grid_for(rows * WAVE, 256)), dim3(256),
                     0, stream.stream(), lf.data_ptr<float>(),
                     lab.data_ptr<long>(), probs.data_ptr<float>(),
                     rowloss.data_ptr<float>(), T, M, ways);
  hipLaunchKernelGGL(ce_rowmean_kernel, dim3((T + 255) / 256), dim3(256), 0,
                     stream.stream(), rowloss.data_ptr<float>(),
                     loss.data_ptr<float>(), T, M);
  return {loss, probs};
}

std::vector<torch::Tensor> ce_dbwd(torch::Tensor probs, torch::Tensor labels,
                                   torch::Tensor gdl, torch::Tensor gtask) {
  const int T = (int)probs.size(0), M = (int)probs.size(1),
            ways = (int)probs.size(2);
  auto d_logits = torch::empty_like(probs);
  auto d_gtask = torch::empty({T}, probs.options());
  auto d_gtask_rows = torch::empty({T, M}, probs.options());
  auto lab = labels.contiguous().to(torch::kLong);
  auto gdlc = gdl.contiguous().to(torch::kFloat32);
  auto gtc = gtask.contiguous().to(torch::kFloat32);
  auto stream = at::cuda::getCurrentCUDAStream();
  const long rows = (long)T * M;
  hipLaunchKernelGGL(ce_dbwd_kernel, dim3(grid_for(rows * WAVE, 256)),
                     dim3(256), 0, stream.stream(), probs.data_ptr<float>(),
                     lab.data_ptr<long>(), gdlc.data_ptr<float>(),
                     gtc.data_ptr<float>(), d_logits.data_ptr<float>(),
                     d_gtask_rows.data_ptr<float>(), T, M, ways);
  hipLaunchKernelGGL(ce_rowmean_kernel, dim3((T + 255) / 256), dim3(256), 0,
                     stream.stream(), d_gtask_rows.data_ptr<float>(),
                     d_gtask.data_ptr<float>(), T, M);
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
