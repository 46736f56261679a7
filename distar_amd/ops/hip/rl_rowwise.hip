// Fused rowwise entropy / teacher-KL over large class dims (SURVEY §2.9
// K13, completing ce_loss.hip).
//
// RL loss shapes (rl_utils.py entropy_loss/kl_loss): rows = T*B (policy
// heads) with C up to 24320 (location).  Eager materializes softmax AND
// log_softmax (N,C) for entropy, and two log_softmax tensors for KL; these
// kernels keep everything in registers/LDS and write only per-row scalars
// forward (+ one (N,C) grad tensor backward).
//
// entropy:  H_i  = lse_i - sum_j p_ij l_ij;          dH/dl_k = -p_k (log p_k + H)
// kl(T||S): kl_i = sum_j pT_ij (logpT_ij - logpS_ij); d/dlS_k = pS_k - pT_k
//
// Layout: one 256-thread workgroup per row (4 waves), threads stride the
// row; block reductions through LDS as in ce_loss.hip.
//
// EXPERIMENTAL (round-2 validation pending): compiled and bound, enabled
// only via DISTAR_AMD_FUSED_RL_ROWWISE=1.
#include <hip/hip_runtime.h>
#include <cfloat>

#define RW_NT 256

__device__ inline float rw_block_max(float v, float* lds) {
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_down(v, off, 64));
  int wave = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) lds[wave] = v;
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < RW_NT / 64; ++w) v = fmaxf(v, lds[w]);
    lds[0] = v;
  }
  __syncthreads();
  v = lds[0];
  __syncthreads();
  return v;
}

__device__ inline float rw_block_sum(float v, float* lds) {
  for (int off = 32; off > 0; off >>= 1)
    v += __shfl_down(v, off, 64);
  int wave = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) lds[wave] = v;
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < RW_NT / 64; ++w) v += lds[w];
    lds[0] = v;
  }
  __syncthreads();
  v = lds[0];
  __syncthreads();
  return v;
}

// H_i = lse_i - sum p*l ; lse saved for backward.
extern "C" __global__ void entropy_fwd_kernel(
    const float* __restrict__ logits,   // (N, C)
    float* __restrict__ entropy,        // (N,)
    float* __restrict__ lse,            // (N,)
    int N, int C) {
  __shared__ float lds[RW_NT / 64];
  int row = blockIdx.x;
  if (row >= N) return;
  const float* l = logits + (size_t)row * C;
  float m = -FLT_MAX;
  for (int j = threadIdx.x; j < C; j += blockDim.x)
    m = fmaxf(m, l[j]);
  m = rw_block_max(m, lds);
  float s = 0.f, sl = 0.f;
  for (int j = threadIdx.x; j < C; j += blockDim.x) {
    float e = __expf(l[j] - m);
    s += e;
    sl += e * l[j];
  }
  s = rw_block_sum(s, lds);
  sl = rw_block_sum(sl, lds);
  if (threadIdx.x == 0) {
    float L = m + __logf(s);
    lse[row] = L;
    entropy[row] = L - sl / s;          // lse - E_p[l]
  }
}

// dH/dl_k = -p_k (log p_k + H) * gout_i
extern "C" __global__ void entropy_bwd_kernel(
    const float* __restrict__ logits,   // (N, C)
    const float* __restrict__ lse,      // (N,)
    const float* __restrict__ entropy,  // (N,)
    const float* __restrict__ gout,     // (N,)
    float* __restrict__ dlogits,        // (N, C)
    int N, int C) {
  int row = blockIdx.x;
  if (row >= N) return;
  const float* l = logits + (size_t)row * C;
  float* d = dlogits + (size_t)row * C;
  float L = lse[row], H = entropy[row], g = gout[row];
  for (int j = threadIdx.x; j < C; j += blockDim.x) {
    float lp = l[j] - L;                // log p
    d[j] = -g * __expf(lp) * (lp + H);
  }
}

// kl_i = sum pT (logpT - logpS); both lse's saved for backward.
extern "C" __global__ void kl_fwd_kernel(
    const float* __restrict__ t_logits, // (N, C) teacher (no grad)
    const float* __restrict__ s_logits, // (N, C) student
    float* __restrict__ kl,             // (N,)
    float* __restrict__ t_lse,          // (N,)
    float* __restrict__ s_lse,          // (N,)
    int N, int C) {
  __shared__ float lds[RW_NT / 64];
  int row = blockIdx.x;
  if (row >= N) return;
  const float* lt = t_logits + (size_t)row * C;
  const float* ls = s_logits + (size_t)row * C;
  float mt = -FLT_MAX, ms = -FLT_MAX;
  for (int j = threadIdx.x; j < C; j += blockDim.x) {
    mt = fmaxf(mt, lt[j]);
    ms = fmaxf(ms, ls[j]);
  }
  mt = rw_block_max(mt, lds);
  ms = rw_block_max(ms, lds);
  float st = 0.f, ss = 0.f;
  for (int j = threadIdx.x; j < C; j += blockDim.x) {
    st += __expf(lt[j] - mt);
    ss += __expf(ls[j] - ms);
  }
  st = rw_block_sum(st, lds);
  ss = rw_block_sum(ss, lds);
  float Lt = mt + __logf(st), Ls = ms + __logf(ss);
  float acc = 0.f;
  for (int j = threadIdx.x; j < C; j += blockDim.x) {
    float lpt = lt[j] - Lt;
    acc += __expf(lpt) * (lpt - (ls[j] - Ls));
  }
  acc = rw_block_sum(acc, lds);
  if (threadIdx.x == 0) {
    kl[row] = acc;
    t_lse[row] = Lt;
    s_lse[row] = Ls;
  }
}

// d kl / d lS_k = (pS_k - pT_k) * gout_i
extern "C" __global__ void kl_bwd_kernel(
    const float* __restrict__ t_logits,
    const float* __restrict__ s_logits,
    const float* __restrict__ t_lse,
    const float* __restrict__ s_lse,
    const float* __restrict__ gout,
    float* __restrict__ ds_logits,      // (N, C)
    int N, int C) {
  int row = blockIdx.x;
  if (row >= N) return;
  const float* lt = t_logits + (size_t)row * C;
  const float* ls = s_logits + (size_t)row * C;
  float* d = ds_logits + (size_t)row * C;
  float Lt = t_lse[row], Ls = s_lse[row], g = gout[row];
  for (int j = threadIdx.x; j < C; j += blockDim.x)
    d[j] = g * (__expf(ls[j] - Ls) - __expf(lt[j] - Lt));
}
