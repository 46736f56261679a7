// Selected-units pointer-network SAMPLING loop as one persistent kernel
// (SURVEY §2.9 K7 — "the latency-critical inference kernel").
//
// Eager runs up to 64 data-dependent sequential steps, each ~15 kernel
// launches (two query fcs, an LN-LSTM cell, a key dot, masked softmax,
// multinomial, two embed fcs).  Here ONE workgroup per batch row executes
// its whole loop: keys staged in LDS once (513 x 32 bf16 = 32 KB), the
// small weight matrices streamed from L2 each step, LayerNorms and softmax
// as block reductions, and sampling via inverse-CDF against HOST-drawn
// uniforms (torch RNG => reproducible, golden-testable).
//
// Semantics follow `action_arg_head.py:262-313`: end flag masked at step 0
// and enabled from step 1, previously-selected units masked, running
// mean-of-selected-keys added back through the embed fcs, per-row stop at
// the end token (the reference only stops globally; steps past a row's end
// are loss-masked, so per-row stop is equivalent on everything consumed).
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define SNT 256          // threads per block (4 waves)
#define MAX_SEL 64
#define KEY_DIM 32
#define LN_EPS 1e-5f

using bf16 = __hip_bfloat16;

__device__ inline float bf2f(bf16 v) { return __bfloat162float(v); }

__device__ inline float block_sum_s(float v, float* scratch) {
  int tid = threadIdx.x;
  scratch[tid] = v;
  __syncthreads();
  for (int s = SNT / 2; s > 0; s >>= 1) {
    if (tid < s) scratch[tid] += scratch[tid + s];
    __syncthreads();
  }
  float r = scratch[0];
  __syncthreads();
  return r;
}

__device__ inline float block_max_s(float v, float* scratch) {
  int tid = threadIdx.x;
  scratch[tid] = v;
  __syncthreads();
  for (int s = SNT / 2; s > 0; s >>= 1) {
    if (tid < s) scratch[tid] = fmaxf(scratch[tid], scratch[tid + s]);
    __syncthreads();
  }
  float r = scratch[0];
  __syncthreads();
  return r;
}

// y[j] = act(b[j] + sum_k x[k] * W[j,k]) for j in [0, OUT); W row-major
// (OUT, K) bf16, x fp32 in LDS.  act: 0 = none, 1 = relu.
__device__ inline void fc_lds(const float* __restrict__ x, int K,
                              const bf16* __restrict__ W,
                              const float* __restrict__ b, int OUT,
                              float* __restrict__ y, int act) {
  for (int j = threadIdx.x; j < OUT; j += SNT) {
    const bf16* wrow = W + (long)j * K;
    float acc = b ? b[j] : 0.f;
    for (int k = 0; k < K; k += 2) {
      acc = fmaf(x[k], bf2f(wrow[k]), acc);
      acc = fmaf(x[k + 1], bf2f(wrow[k + 1]), acc);
    }
    y[j] = act == 1 ? fmaxf(acc, 0.f) : acc;
  }
  __syncthreads();
}

extern "C" __global__ void su_sample_kernel(
    const float* __restrict__ ae_base,    // (B, 1024)
    const bf16* __restrict__ keys,        // (B, N1, 32)  N1 = N + 1 end slot
    const unsigned char* __restrict__ avail,  // (B, N1) availability
    const unsigned char* __restrict__ su_mask, // (B)
    const int* __restrict__ entity_num,   // (B)
    const float* __restrict__ uniforms,   // (B, MAX_SEL)
    // weights (bf16 rows, fp32 bias / LN params)
    const bf16* __restrict__ Wq1, const float* __restrict__ bq1,  // (256,1024)
    const bf16* __restrict__ Wq2, const float* __restrict__ bq2,  // (32,256)
    const bf16* __restrict__ Wih, const bf16* __restrict__ Whh,   // (128,32)
    const float* __restrict__ lni_w, const float* __restrict__ lni_b,   // 128
    const float* __restrict__ lnh_w, const float* __restrict__ lnh_b,   // 128
    const float* __restrict__ lnc_w, const float* __restrict__ lnc_b,   // 32
    const bf16* __restrict__ We1, const float* __restrict__ be1,  // (256,32)
    const bf16* __restrict__ We2, const float* __restrict__ be2,  // (1024,256)
    float temperature,
    float* __restrict__ logits_out,       // (B, MAX_SEL, N1) pre-filled -1e9
    int* __restrict__ results,            // (B, MAX_SEL)
    int* __restrict__ num_out,            // (B)
    int B, int N1, int AE) {              // AE = 1024
  const int row = blockIdx.x;
  const int tid = threadIdx.x;
  if (row >= B) return;
  extern __shared__ float smem[];
  float* ae = smem;                 // AE
  float* q1 = ae + AE;              // 256
  float* q2 = q1 + 256;             // 32
  float* gates = q2 + 32;           // 128
  float* hst = gates + 128;         // 32
  float* cst = hst + 32;            // 32
  float* selsum = cst + 32;         // 32
  float* emb1 = selsum + 32;        // 256
  float* logit = emb1 + 256;        // N1 (runtime)
  float* scratch = logit + N1;      // SNT
  unsigned char* mask = reinterpret_cast<unsigned char*>(scratch + SNT);  // N1
  bf16* keyl = reinterpret_cast<bf16*>(
      reinterpret_cast<char*>(mask) + ((N1 + 63) / 64) * 64);  // N1*32 bf16

  const int en = entity_num[row];
  if (!su_mask[row]) {
    if (tid == 0) num_out[row] = 0;
    return;
  }
  // stage row state
  for (int j = tid; j < AE; j += SNT) ae[j] = ae_base[(long)row * AE + j];
  for (int n = tid; n < N1; n += SNT) mask[n] = avail[(long)row * N1 + n];
  for (int j = tid; j < N1 * KEY_DIM; j += SNT)
    keyl[j] = keys[(long)row * N1 * KEY_DIM + j];
  for (int j = tid; j < 32; j += SNT) {
    hst[j] = 0.f;
    cst[j] = 0.f;
    selsum[j] = 0.f;
  }
  __syncthreads();
  if (tid == 0) mask[en] = 0;       // end flag unavailable at step 0
  __syncthreads();

  int num = MAX_SEL;
  float selcnt = 0.f;
  for (int i = 0; i < MAX_SEL; ++i) {
    if (i == 1) {
      if (tid == 0 && en < N1) mask[en] = avail[(long)row * N1 + en];
      __syncthreads();
    }
    // query fcs
    fc_lds(ae, AE, Wq1, bq1, 256, q1, 1);
    fc_lds(q1, 256, Wq2, bq2, 32, q2, 0);
    // LN-LSTM cell (H=32, G=128)
    fc_lds(q2, 32, Wih, nullptr, 128, gates, 0);       // igates_raw -> gates
    {
      float ls = 0.f, lq = 0.f;
      for (int g = tid; g < 128; g += SNT) { ls += gates[g]; lq += gates[g] * gates[g]; }
      float mean = block_sum_s(ls, scratch) / 128.f;
      float var = block_sum_s(lq, scratch) / 128.f - mean * mean;
      float rstd = rsqrtf(var + LN_EPS);
      for (int g = tid; g < 128; g += SNT)
        gates[g] = (gates[g] - mean) * rstd * lni_w[g] + lni_b[g];
      __syncthreads();
      // hgates into q1[0..128) as scratch
      fc_lds(hst, 32, Whh, nullptr, 128, q1, 0);
      ls = 0.f; lq = 0.f;
      for (int g = tid; g < 128; g += SNT) { ls += q1[g]; lq += q1[g] * q1[g]; }
      mean = block_sum_s(ls, scratch) / 128.f;
      var = block_sum_s(lq, scratch) / 128.f - mean * mean;
      rstd = rsqrtf(var + LN_EPS);
      for (int g = tid; g < 128; g += SNT)
        gates[g] += (q1[g] - mean) * rstd * lnh_w[g] + lnh_b[g];
      __syncthreads();
      // cell update
      for (int k = tid; k < 32; k += SNT) {
        float ig = 1.f / (1.f + expf(-gates[k]));
        float fg = 1.f / (1.f + expf(-gates[k + 32]));
        float gg = tanhf(gates[k + 64]);
        cst[k] = fg * cst[k] + ig * gg;     // raw cell, LN next
      }
      __syncthreads();
      ls = 0.f; lq = 0.f;
      for (int k = tid; k < 32; k += SNT) { ls += cst[k]; lq += cst[k] * cst[k]; }
      mean = block_sum_s(ls, scratch) / 32.f;
      var = block_sum_s(lq, scratch) / 32.f - mean * mean;
      rstd = rsqrtf(var + LN_EPS);
      for (int k = tid; k < 32; k += SNT) {
        float og = 1.f / (1.f + expf(-gates[k + 96]));
        cst[k] = (cst[k] - mean) * rstd * lnc_w[k] + lnc_b[k];
        hst[k] = og * tanhf(cst[k]);
      }
      __syncthreads();
    }
    // key dot + mask + temperature
    for (int n = tid; n < N1; n += SNT) {
      const bf16* krow = keyl + n * KEY_DIM;
      float acc = 0.f;
#pragma unroll
      for (int k = 0; k < KEY_DIM; ++k) acc = fmaf(hst[k], bf2f(krow[k]), acc);
      logit[n] = mask[n] ? acc / temperature : -1e9f;
    }
    __syncthreads();
    // softmax + inverse-CDF sample
    float lmax = -1e30f;
    for (int n = tid; n < N1; n += SNT) lmax = fmaxf(lmax, logit[n]);
    lmax = block_max_s(lmax, scratch);
    float lsum = 0.f;
    for (int n = tid; n < N1; n += SNT) lsum += expf(logit[n] - lmax);
    lsum = block_sum_s(lsum, scratch);
    if (tid == 0) {
      float target = uniforms[(long)row * MAX_SEL + i] * lsum;
      float cdf = 0.f;
      int pick = -1;
      for (int n = 0; n < N1; ++n) {
        cdf += expf(logit[n] - lmax);
        if (cdf >= target) { pick = n; break; }
      }
      if (pick < 0) {               // numeric tail: last unmasked index
        for (int n = N1 - 1; n >= 0; --n) if (mask[n]) { pick = n; break; }
        if (pick < 0) pick = en;
      }
      scratch[0] = (float)pick;
    }
    __syncthreads();
    int pick = (int)scratch[0];
    // emit logits + result
    for (int n = tid; n < N1; n += SNT)
      logits_out[((long)row * MAX_SEL + i) * N1 + n] = logit[n];
    if (tid == 0) results[(long)row * MAX_SEL + i] = pick;
    __syncthreads();
    if (pick == en) { num = i + 1; break; }
    // mask the pick, update mean-of-selected embedding feedback
    if (tid == 0) mask[pick] = 0;
    for (int k = tid; k < 32; k += SNT)
      selsum[k] += bf2f(keyl[pick * KEY_DIM + k]);
    selcnt += 1.f;
    __syncthreads();
    for (int k = tid; k < 32; k += SNT) q2[k] = selsum[k] / selcnt;  // mean
    __syncthreads();
    fc_lds(q2, 32, We1, be1, 256, emb1, 1);
    // ae = ae_base + We2 @ emb1
    for (int j = tid; j < AE; j += SNT) {
      const bf16* wrow = We2 + (long)j * 256;
      float acc = be2[j];
      for (int k = 0; k < 256; k += 2) {
        acc = fmaf(emb1[k], bf2f(wrow[k]), acc);
        acc = fmaf(emb1[k + 1], bf2f(wrow[k + 1]), acc);
      }
      ae[j] = ae_base[(long)row * AE + j] + acc;
    }
    __syncthreads();
  }
  if (tid == 0) num_out[row] = num;
}
