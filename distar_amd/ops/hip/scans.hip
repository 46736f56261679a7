// (T, B) reverse scans for the RL losses — V-trace corrected values and
// generalized lambda returns.
//
// Sequential in T, parallel in B: one lane per batch column, grid-stride
// over columns.  Row layout is (T, B) contiguous, so at each timestep the
// wave's 64 lanes read 64 consecutive floats — fully coalesced 256 B
// transactions per row.  T <= 128, B = batch x heads x baselines; the whole
// scan is one kernel launch instead of the reference's T-iteration Python
// loop over torch ops (as_rl_utils.py:157-218,284-312).
#include <hip/hip_runtime.h>

extern "C" __global__ void lambda_return_kernel(
    const float* __restrict__ rewards,       // (T, B)
    const float* __restrict__ gammas,        // (T, B)
    const float* __restrict__ values_tp1,    // (T, B)  V_{t+1}
    const float* __restrict__ lambdas,       // (T, B)
    float* __restrict__ out,                 // (T, B)
    int T, int B) {
  for (int b = blockIdx.x * blockDim.x + threadIdx.x; b < B;
       b += gridDim.x * blockDim.x) {
    int i = (T - 1) * B + b;
    float acc = rewards[i] + gammas[i] * values_tp1[i];
    out[i] = acc;
    for (int t = T - 2; t >= 0; --t) {
      i = t * B + b;
      float disc = gammas[i] * lambdas[i];
      acc = rewards[i] + disc * acc + (gammas[i] - disc) * values_tp1[i];
      out[i] = acc;
    }
  }
}

extern "C" __global__ void vtrace_kernel(
    const float* __restrict__ clipped_rhos,  // (T, B)
    const float* __restrict__ clipped_cs,    // (T, B)
    const float* __restrict__ rewards,       // (T, B)
    const float* __restrict__ values,        // (T+1, B)
    const float* __restrict__ gammas,        // (T, B)
    const float* __restrict__ lambdas,       // (T, B)
    float* __restrict__ vtrace_out,          // (T+1, B)
    int T, int B) {
  for (int b = blockIdx.x * blockDim.x + threadIdx.x; b < B;
       b += gridDim.x * blockDim.x) {
    float v_next = values[T * B + b];         // V_T
    float vt_next = v_next;                   // vtrace_{T}
    vtrace_out[T * B + b] = vt_next;
    for (int t = T - 1; t >= 0; --t) {
      int i = t * B + b;
      float v_t = values[i];
      float delta = clipped_rhos[i] * (rewards[i] + gammas[i] * v_next - v_t);
      float vt = v_t + delta +
          gammas[i] * lambdas[i] * clipped_cs[i] * (vt_next - v_next);
      vtrace_out[i] = vt;
      vt_next = vt;
      v_next = v_t;
    }
  }
}
