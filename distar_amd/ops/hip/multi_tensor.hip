// K14: multi-tensor global-norm + scale kernels (grad clipping).
//
// The reference clips with per-parameter torch ops (`grad_clip.py:35-151`);
// round 1 used torch's foreach paths.  These kernels do the whole
// parameter set in two launches driven by a (ptr, numel) chunk table:
//   multi_norm_sq: sum of squares over every chunk -> one fp32 scalar
//   multi_scale:   p *= *scale (device scalar — no host sync in the loop)
#include <hip/hip_runtime.h>

extern "C" __global__ void multi_norm_sq_kernel(
    const unsigned long long* __restrict__ ptrs,
    const long* __restrict__ numels,
    int n_chunks, float* __restrict__ out) {
  float acc = 0.f;
  for (int c = blockIdx.x; c < n_chunks; c += gridDim.x) {
    const float* p = reinterpret_cast<const float*>(ptrs[c]);
    long n = numels[c];
    for (long i = threadIdx.x * 4; i + 3 < n; i += (long)blockDim.x * 4) {
      float4 v = *reinterpret_cast<const float4*>(p + i);   // 16B loads
      acc += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
    }
    if (threadIdx.x == 0)                 // scalar tail (n % 4 elements)
      for (long t = n - (n & 3); t < n; ++t) acc += p[t] * p[t];
  }
  // block reduce
  __shared__ float warp_sums[4];
  for (int off = 1; off < 64; off <<= 1)
    acc += __shfl_xor(acc, off, 64);
  int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  if (lane == 0) warp_sums[wave] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float s = 0.f;
    for (int w = 0; w < (int)(blockDim.x >> 6); ++w) s += warp_sums[w];
    atomicAdd(out, s);
  }
}

// scale_mode 0: p *= *scale
// scale_mode 1: p *= min(1, thresh / (sqrt(*norm_sq) + eps))  (clip-by-norm)
extern "C" __global__ void multi_scale_kernel(
    const unsigned long long* __restrict__ ptrs,
    const long* __restrict__ numels,
    int n_chunks, const float* __restrict__ norm_sq,
    float thresh, float eps, int scale_mode) {
  float s;
  if (scale_mode == 1) {
    float norm = sqrtf(*norm_sq);
    s = thresh / (norm + eps);
    if (s > 1.f) s = 1.f;
  } else {
    s = *norm_sq;                       // raw scale passed through slot 0
  }
  for (int c = blockIdx.y; c < n_chunks; c += gridDim.y) {
    float* p = reinterpret_cast<float*>(ptrs[c]);
    long n = numels[c];
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x)
      p[i] *= s;
  }
}
