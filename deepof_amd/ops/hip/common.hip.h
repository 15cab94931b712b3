// Shared device helpers for deepof_amd CDNA4 (gfx950) kernels.
// Wavefront = 64 lanes; block sizes are multiples of 64.
#pragma once

#include <hip/hip_runtime.h>

#define DEEPOF_WAVE 64

// Wave-then-LDS block sum; the RESULT IS VALID IN THREAD 0 ONLY.
// `lds` must hold blockDim.x/64 floats.
__device__ inline float block_reduce_sum(float val, float* lds) {
  const int lane = threadIdx.x & (DEEPOF_WAVE - 1);
  const int wid = threadIdx.x >> 6;
#pragma unroll
  for (int off = DEEPOF_WAVE / 2; off > 0; off >>= 1)
    val += __shfl_down(val, off, DEEPOF_WAVE);
  if (lane == 0) lds[wid] = val;
  __syncthreads();
  const int nwaves = blockDim.x >> 6;
  val = (threadIdx.x < nwaves) ? lds[threadIdx.x] : 0.0f;
  if (wid == 0) {
#pragma unroll
    for (int off = DEEPOF_WAVE / 2; off > 0; off >>= 1)
      val += __shfl_down(val, off, DEEPOF_WAVE);
  }
  return val;
}

// (d^2 + eps^2)^alpha; alpha = 0.25 (the photometric default) is two
// exact sqrtfs instead of a powf.
__device__ inline float charb(float d, float eps2, float alpha) {
  const float t = d * d + eps2;
  if (alpha == 0.25f) return sqrtf(sqrtf(t));
  if (alpha == 0.5f) return sqrtf(t);
  return powf(t, alpha);
}

// d/dd of (d^2 + eps^2)^alpha = alpha * (d^2+eps^2)^(alpha-1) * 2d
__device__ inline float charb_grad(float d, float eps2, float alpha) {
  const float t = d * d + eps2;
  if (alpha == 0.25f) return 0.5f * d * sqrtf(sqrtf(t)) / t;
  return alpha * powf(t, alpha - 1.0f) * 2.0f * d;
}

__device__ inline bool in_border(int y, int x, int h, int w, int bw) {
  return y >= bw && y < h - bw && x >= bw && x < w - bw;
}

#define DEEPOF_CHECK_HIP(expr)                                         \
  do {                                                                 \
    hipError_t _e = (expr);                                            \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e)); \
  } while (0)
