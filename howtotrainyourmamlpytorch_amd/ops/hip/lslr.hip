// Fused LSLR fast-weight update over the flat arena:
//   out[t, p] = arena[t, p] - lr_vec[p] * grad[t, p]
// One elementwise kernel for the entire task batch's update — the
// reference's per-tensor Python loop (inner_loop_optimizers.py:99-113)
// collapsed into a single launch.  The backward kernel produces
// d_grad = -lr_vec * gout  and  d_lr_vec[p] = -sum_t gout[t,p] * grad[t,p]
// in one fused pass.

#include "common.h"
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

using namespace maml355;

__global__ void lslr_fwd_kernel(const float* __restrict__ arena,
                                const float* __restrict__ grad,
                                const float* __restrict__ lr_vec,
                                float* __restrict__ out, int T, long P) {
  const long total = (long)T * P;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += grid_stride()) {
    const long p = i % P;
    out[i] = fmaf(-lr_vec[p], grad[i], arena[i]);
  }
}

// one thread per arena column p: loops over the (small) task dim, emitting
// d_grad and the task-reduced d_lr in a single pass.
__global__ void lslr_bwd_kernel(const float* __restrict__ gout,
                                const float* __restrict__ grad,
                                const float* __restrict__ lr_vec,
                                float* __restrict__ dgrad,
                                float* __restrict__ dlr_vec, int T, long P) {
  for (long p = (long)blockIdx.x * blockDim.x + threadIdx.x; p < P;
       p += grid_stride()) {
    const float lr = lr_vec[p];
    float acc = 0.f;
    for (int t = 0; t < T; ++t) {
      const long i = (long)t * P + p;
      const float g = gout[i];
      dgrad[i] = -lr * g;
      acc += g * grad[i];
    }
    dlr_vec[p] = -acc;
  }
}

static int grid_for(long total, int threads) {
  long blocks = (total + threads - 1) / threads;
  return (int)std::min<long>(blocks, 4096);
}

torch::Tensor lslr_fwd(torch::Tensor arena, torch::Tensor grad,
                       torch::Tensor lr_vec) {
  TORCH_CHECK(arena.is_cuda() && arena.dim() == 2 && arena.is_contiguous());
  const int T = (int)arena.size(0);
  const long P = arena.size(1);
  auto out = torch::empty_like(arena);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(lslr_fwd_kernel, dim3(grid_for(T * P, 256)), dim3(256), 0,
                     stream.stream(), arena.data_ptr<float>(),
                     grad.contiguous().data_ptr<float>(),
                     lr_vec.contiguous().data_ptr<float>(),
                     out.data_ptr<float>(), T, P);
  return out;
}

std::vector<torch::Tensor> lslr_bwd(torch::Tensor gout, torch::Tensor grad,
                                    torch::Tensor lr_vec) {
  const int T = (int)gout.size(0);
  const long P = gout.size(1);
  auto goutc = gout.contiguous();
  auto gradc = grad.contiguous();
  auto dgrad = torch::empty_like(goutc);
  auto dlr = torch::empty({P}, gout.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(lslr_bwd_kernel, dim3(grid_for(P, 256)), dim3(256), 0,
                     stream.stream(), goutc.data_ptr<float>(),
                     gradc.data_ptr<float>(),
                     lr_vec.contiguous().data_ptr<float>(),
                     dgrad.data_ptr<float>(), dlr.data_ptr<float>(), T, P);
  return {dgrad, dlr};
}
