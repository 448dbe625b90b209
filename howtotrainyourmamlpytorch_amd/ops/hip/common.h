// Shared helpers for the MI355X (gfx950 / CDNA4) kernels.
// Wave width is 64 on CDNA — hard-coded per the platform guide.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define DEVINL __device__ __forceinline__

namespace maml355 {

DEVINL float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    v += __shfl_down(v, off, WAVE);
  }
  return v;
}

DEVINL float wave_reduce_max(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    v = fmaxf(v, __shfl_down(v, off, WAVE));
  }
  return v;
}

// Block-level sum reduction into lane 0 of wave 0. `lds` must hold
// >= blockDim.x / WAVE floats.
DEVINL float block_reduce_sum(float v, float* lds) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  const int nwaves = blockDim.x / WAVE;
  v = (threadIdx.x < nwaves) ? lds[threadIdx.x] : 0.f;
  if (wid == 0) v = wave_reduce_sum(v);
  return v;
}

// grid-stride loop bound: cap resident blocks, stride the rest (guide G11)
DEVINL long grid_stride() { return (long)blockDim.x * gridDim.x; }

template <typename T>
DEVINL float to_f32(T v);
template <>
DEVINL float to_f32<float>(float v) { return v; }
template <>
DEVINL float to_f32<__hip_bfloat16>(__hip_bfloat16 v) { return __bfloat162float(v); }

template <typename T>
DEVINL T from_f32(float v);
template <>
DEVINL float from_f32<float>(float v) { return v; }
template <>
DEVINL __hip_bfloat16 from_f32<__hip_bfloat16>(float v) { return __float2bfloat16(v); }

// 8-wide vector load/store (16B for bf16, 32B for fp32) — guide G13:
// hipcc does not auto-vectorize scalar bf16 loads.
typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8_t;

template <typename T>
DEVINL void load8(const T* p, float* out);
template <>
DEVINL void load8<float>(const float* p, float* out) {
#pragma unroll
  for (int j = 0; j < 8; ++j) out[j] = p[j];
}
template <>
DEVINL void load8<__hip_bfloat16>(const __hip_bfloat16* p, float* out) {
  const bf16x8_t v = *(const bf16x8_t*)p;
#pragma unroll
  for (int j = 0; j < 8; ++j)
    out[j] = __bfloat162float(
        __hip_bfloat16(__hip_bfloat16_raw{(unsigned short)v[j]}));
}

template <typename T>
DEVINL void store8(T* p, const float* in);
template <>
DEVINL void store8<float>(float* p, const float* in) {
#pragma unroll
  for (int j = 0; j < 8; ++j) p[j] = in[j];
}
template <>
DEVINL void store8<__hip_bfloat16>(__hip_bfloat16* p, const float* in) {
  bf16x8_t v;
#pragma unroll
  for (int j = 0; j < 8; ++j)
    v[j] = (short)__hip_bfloat16_raw(__float2bfloat16(in[j])).x;
  *(bf16x8_t*)p = v;
}

}  // namespace maml355
