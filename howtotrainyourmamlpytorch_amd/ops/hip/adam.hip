// Fused multi-tensor Adam step + gradient clamp for the meta-update
// (reference: few_shot_learning_system.py:330-336 — per-tensor
// grad.clamp_(-10,10) followed by torch Adam over every meta-parameter).
//
// One kernel launch updates every trainable tensor: the tensor table
// (pointers + prefix offsets) is passed by value in the kernel-arg block,
// each thread locates its tensor with a register scan (<= 32 tensors,
// ~126k total elements for the flagship configs — launch-latency bound,
// so a single fused launch replaces torch's ~10 foreach launches).
// fp32 params/state only (the arena/master weights are fp32 by design).

#include "common.h"
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

using namespace maml355;

#define ADAM_MAX_TENSORS 32

struct AdamTable {
  float* p[ADAM_MAX_TENSORS];
  float* g[ADAM_MAX_TENSORS];
  float* m[ADAM_MAX_TENSORS];
  float* v[ADAM_MAX_TENSORS];
  long offset[ADAM_MAX_TENSORS + 1];  // prefix sums; offset[n] = total
  int n;
};

__global__ void fused_adam_kernel(AdamTable tab, float lr, float beta1,
                                  float beta2, float eps, float weight_decay,
                                  float clamp_v, float bias_c1, float bias_c2) {
  const long total = tab.offset[tab.n];
  const float inv_bc1 = 1.f / bias_c1;
  const float inv_sqrt_bc2 = rsqrtf(bias_c2);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += grid_stride()) {
    int t = 0;
#pragma unroll
    for (int k = 0; k < ADAM_MAX_TENSORS; ++k) {
      if (k < tab.n && i >= tab.offset[k + 1]) t = k + 1;
    }
    const long j = i - tab.offset[t];
    float g = tab.g[t][j];
    if (clamp_v > 0.f) g = fminf(fmaxf(g, -clamp_v), clamp_v);
    float p = tab.p[t][j];
    if (weight_decay != 0.f) g += weight_decay * p;
    float m = beta1 * tab.m[t][j] + (1.f - beta1) * g;
    float v = beta2 * tab.v[t][j] + (1.f - beta2) * g * g;
    tab.m[t][j] = m;
    tab.v[t][j] = v;
    const float denom = sqrtf(v) * inv_sqrt_bc2 + eps;
    tab.p[t][j] = p - lr * inv_bc1 * m / denom;
  }
}

// params/grads/exp_avgs/exp_avg_sqs: equal-length lists of fp32 CUDA
// tensors; step is the POST-increment step count (t >= 1).
void adam_step(std::vector<torch::Tensor> params,
               std::vector<torch::Tensor> grads,
               std::vector<torch::Tensor> exp_avgs,
               std::vector<torch::Tensor> exp_avg_sqs,
               long step, double lr, double beta1, double beta2, double eps,
               double weight_decay, double clamp_v) {
  const int n = (int)params.size();
  TORCH_CHECK(n > 0 && n <= ADAM_MAX_TENSORS,
              "adam_step supports 1..32 tensors, got ", n);
  AdamTable tab;
  tab.n = n;
  long off = 0;
  for (int i = 0; i < n; ++i) {
    TORCH_CHECK(params[i].is_cuda() && params[i].is_contiguous() &&
                    params[i].scalar_type() == torch::kFloat32,
                "adam_step wants contiguous fp32 CUDA params");
    TORCH_CHECK(grads[i].numel() == params[i].numel());
    tab.p[i] = params[i].data_ptr<float>();
    tab.g[i] = grads[i].data_ptr<float>();
    tab.m[i] = exp_avgs[i].data_ptr<float>();
    tab.v[i] = exp_avg_sqs[i].data_ptr<float>();
    tab.offset[i] = off;
    off += params[i].numel();
  }
  tab.offset[n] = off;
  const float bc1 = 1.f - powf((float)beta1, (float)step);
  const float bc2 = 1.f - powf((float)beta2, (float)step);
  const int threads = 256;
  const int blocks = (int)std::min<long>((off + threads - 1) / threads, 4096);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(fused_adam_kernel, dim3(blocks), dim3(threads), 0,
                     stream.stream(), tab, (float)lr, (float)beta1,
                     (float)beta2, (float)eps, (float)weight_decay,
                     (float)clamp_v, bc1, bc2);
}
