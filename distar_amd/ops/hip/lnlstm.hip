// Fused layer-norm LSTM layer: the FULL T-step unroll in one kernel per
// direction.
//
// Semantics match the reference cell (`model/lstm.py:120-153`):
//   igates = LN_i(x @ W_ih^T)            (precomputed outside, one big GEMM)
//   hgates = LN_h(h @ W_hh^T)
//   i,f,g,o = sigmoid/tanh(igates + hgates)
//   c' = LN_c(f*c + i*g);  h' = o * tanh(c')
//
// MI355X design:
//  - one workgroup per batch row.  The recurrence couples timesteps but NOT
//    rows, so row-blocks run the whole unroll independently — zero grid
//    syncs, one launch for all T steps (the eager loop is ~8 launches/step).
//  - W_hh is read as bf16 (matching autocast GEMM precision) and stays
//    L2-resident (1.2 MB for the 384-wide core; every block reads the same
//    rows).  fp32 accumulate.
//  - LayerNorm stats are block-wide LDS tree reductions (4H = 1536 for the
//    core LSTM; 128 for the selected-units pointer LSTM).
//  - backward recomputes LN stats and gate activations from the saved raw
//    gate pre-activations, and emits dgates / d(hgates_raw) so the two big
//    weight-gradient GEMMs (dW_ih, dW_hh) run as single hipBLASLt GEMMs
//    outside the kernel.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define NT 512          // threads per block (8 waves)
#define LN_EPS 1e-5f

using bf16 = __hip_bfloat16;
using bf16x2 = __hip_bfloat162;

__device__ inline float bf2f(bf16 v) { return __bfloat162float(v); }

// dot of 8 bf16 pairs loaded as one 16-byte vector each (G13: hipcc does not
// auto-vectorize scalar bf16 loads; 16 B/lane is the coalescing sweet spot)
__device__ inline float dot8_bf16(const uint4 a, const uint4 b) {
  const bf16x2* pa = reinterpret_cast<const bf16x2*>(&a);
  const bf16x2* pb = reinterpret_cast<const bf16x2*>(&b);
  float acc = 0.f;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    float2 fa = __bfloat1622float2(pa[i]);
    float2 fb = __bfloat1622float2(pb[i]);
    acc = fmaf(fa.x, fb.x, acc);
    acc = fmaf(fa.y, fb.y, acc);
  }
  return acc;
}

// block-wide sum over NT threads; scratch must hold NT floats
__device__ inline float block_sum(float v, float* scratch) {
  int tid = threadIdx.x;
  scratch[tid] = v;
  __syncthreads();
  for (int s = NT / 2; s > 0; s >>= 1) {
    if (tid < s) scratch[tid] += scratch[tid + s];
    __syncthreads();
  }
  float r = scratch[0];
  __syncthreads();
  return r;
}

// ---------------------------------------------------------------- forward
// grid.x = B.  Dynamic LDS layout: [h_bf16: H] [gates: G] [c: H] [scratch: NT]
extern "C" __global__ void lnlstm_forward_kernel(
    const float* __restrict__ igates,    // (T, B, G)   G = 4H, post-LN_i
    const float* __restrict__ h0,        // (B, H)
    const float* __restrict__ c0,        // (B, H)
    const bf16* __restrict__ w_hh,       // (G, H) row-major
    const float* __restrict__ lnh_w, const float* __restrict__ lnh_b,   // (G)
    const float* __restrict__ lnc_w, const float* __restrict__ lnc_b,   // (H)
    float* __restrict__ h_all,           // (T+1, B, H)
    float* __restrict__ c_all,           // (T+1, B, H)
    float* __restrict__ hgates_raw,      // (T, B, G)
    float* __restrict__ cellraw,         // (T, B, H)
    int T, int B, int H) {
  const int row = blockIdx.x;
  const int tid = threadIdx.x;
  const int G = 4 * H;
  extern __shared__ float smem[];
  bf16* hs = reinterpret_cast<bf16*>(smem);            // H bf16
  float* gates = smem + (H + 1) / 2;                   // G floats
  float* c_s = gates + G;                              // H floats
  float* scratch = c_s + H;                            // NT floats

  // init state
  for (int k = tid; k < H; k += NT) {
    float h = h0[row * H + k];
    hs[k] = __float2bfloat16(h);
    c_s[k] = c0[row * H + k];
    h_all[row * H + k] = h;
    c_all[row * H + k] = c_s[k];
  }
  __syncthreads();

  for (int t = 0; t < T; ++t) {
    const float* ig = igates + ((long)t * B + row) * G;
    float* hg_out = hgates_raw + ((long)t * B + row) * G;
    // hgates_j = sum_k bf16(h_k) * W[j,k]
    float local_sum = 0.f, local_sq = 0.f;
    float hg_loc[4];           // up to ceil(G/NT) owned gate columns
    int nown = 0;
    for (int j = tid; j < G; j += NT) {
      const uint4* wrow = reinterpret_cast<const uint4*>(w_hh + (long)j * H);
      const uint4* hv = reinterpret_cast<const uint4*>(hs);
      float acc = 0.f;
      for (int k8 = 0; k8 < H / 8; ++k8) {
        acc += dot8_bf16(hv[k8], wrow[k8]);
      }
      hg_out[j] = acc;
      hg_loc[nown++] = acc;
      local_sum += acc;
      local_sq += acc * acc;
    }
    float mean = block_sum(local_sum, scratch) / G;
    float var = block_sum(local_sq, scratch) / G - mean * mean;
    float rstd = rsqrtf(var + LN_EPS);
    // gates = igates + LN_h(hgates); activation by quadrant
    nown = 0;
    for (int j = tid; j < G; j += NT) {
      float hn = (hg_loc[nown++] - mean) * rstd * lnh_w[j] + lnh_b[j];
      float gate = ig[j] + hn;
      int quad = j / H;
      if (quad == 2) gate = tanhf(gate);                    // cell candidate
      else gate = 1.f / (1.f + expf(-gate));                // i, f, o
      gates[j] = gate;
    }
    __syncthreads();
    // cell update + LN_c + output
    float craw_loc = 0.f;
    float lsum = 0.f, lsq = 0.f;
    for (int k = tid; k < H; k += NT) {
      float i_g = gates[k], f_g = gates[k + H], g_g = gates[k + 2 * H];
      craw_loc = f_g * c_s[k] + i_g * g_g;
      cellraw[((long)t * B + row) * H + k] = craw_loc;
      lsum += craw_loc;
      lsq += craw_loc * craw_loc;
    }
    float cmean = block_sum(lsum, scratch) / H;
    float cvar = block_sum(lsq, scratch) / H - cmean * cmean;
    float crstd = rsqrtf(cvar + LN_EPS);
    for (int k = tid; k < H; k += NT) {
      float craw = cellraw[((long)t * B + row) * H + k];
      float c_new = (craw - cmean) * crstd * lnc_w[k] + lnc_b[k];
      float h_new = gates[k + 3 * H] * tanhf(c_new);
      c_s[k] = c_new;
      hs[k] = __float2bfloat16(h_new);
      h_all[((long)(t + 1) * B + row) * H + k] = h_new;
      c_all[((long)(t + 1) * B + row) * H + k] = c_new;
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------- backward
// grid.x = B.  LDS: [act gates: G] [dgate: G] [dh_rec: H] [dc_rec: H]
//                   [dhg bf-free fp32: G] [scratch: NT]
extern "C" __global__ void lnlstm_backward_kernel(
    const float* __restrict__ dout,      // (T, B, H) grad of h outputs
    const float* __restrict__ dhT,       // (B, H) grad of final h (or null)
    const float* __restrict__ dcT,       // (B, H) grad of final c (or null)
    const float* __restrict__ igates,    // (T, B, G)
    const float* __restrict__ h_all,     // (T+1, B, H)
    const float* __restrict__ c_all,     // (T+1, B, H)
    const float* __restrict__ hgates_raw,// (T, B, G)
    const float* __restrict__ cellraw,   // (T, B, H)
    const bf16* __restrict__ w_hh_t,     // (H, G) = W_hh^T row-major
    const float* __restrict__ lnh_w, const float* __restrict__ lnh_b,
    const float* __restrict__ lnc_w, const float* __restrict__ lnc_b,
    float* __restrict__ digates,         // (T, B, G)
    float* __restrict__ dhgates_raw,     // (T, B, G)  (for dW_hh GEMM)
    float* __restrict__ dh0,             // (B, H)
    float* __restrict__ dc0,             // (B, H)
    float* __restrict__ dlnh_w, float* __restrict__ dlnh_b,   // (G) atomics
    float* __restrict__ dlnc_w, float* __restrict__ dlnc_b,   // (H) atomics
    int T, int B, int H) {
  const int row = blockIdx.x;
  const int tid = threadIdx.x;
  const int G = 4 * H;
  extern __shared__ float smem[];
  float* act = smem;                  // G: activated gates
  float* dgate = act + G;             // G
  float* dhg = dgate + G;             // G
  float* dh_rec = dhg + G;            // H
  float* dc_rec = dh_rec + H;         // H
  float* scratch = dc_rec + H;        // NT

  float dwh_loc[4] = {0, 0, 0, 0}, dbh_loc[4] = {0, 0, 0, 0};
  float dwc_loc = 0.f, dbc_loc = 0.f;

  for (int k = tid; k < H; k += NT) {
    dh_rec[k] = dhT ? dhT[row * H + k] : 0.f;
    dc_rec[k] = dcT ? dcT[row * H + k] : 0.f;
  }
  __syncthreads();

  for (int t = T - 1; t >= 0; --t) {
    const float* ig = igates + ((long)t * B + row) * G;
    const float* hg = hgates_raw + ((long)t * B + row) * G;
    const float* craw = cellraw + ((long)t * B + row) * H;
    const float* c_prev = c_all + ((long)t * B + row) * H;
    const float* c_new = c_all + ((long)(t + 1) * B + row) * H;

    // recompute LN_h stats + gate activations
    float lsum = 0.f, lsq = 0.f;
    for (int j = tid; j < G; j += NT) {
      float v = hg[j];
      lsum += v;
      lsq += v * v;
    }
    float hmean = block_sum(lsum, scratch) / G;
    float hvar = block_sum(lsq, scratch) / G - hmean * hmean;
    float hrstd = rsqrtf(hvar + LN_EPS);
    for (int j = tid; j < G; j += NT) {
      float hn = (hg[j] - hmean) * hrstd * lnh_w[j] + lnh_b[j];
      float gate = ig[j] + hn;
      act[j] = (j / H == 2) ? tanhf(gate) : 1.f / (1.f + expf(-gate));
    }
    // recompute LN_c stats
    lsum = 0.f; lsq = 0.f;
    for (int k = tid; k < H; k += NT) {
      float v = craw[k];
      lsum += v;
      lsq += v * v;
    }
    float cmean = block_sum(lsum, scratch) / H;
    float cvar = block_sum(lsq, scratch) / H - cmean * cmean;
    float crstd = rsqrtf(cvar + LN_EPS);
    __syncthreads();

    // dc_new, LN_c backward, gate grads
    // first pass: compute w_c*dc_new sums for LN_c backward
    float m1_loc = 0.f, m2_loc = 0.f;
    for (int k = tid; k < H; k += NT) {
      float dht = dout[((long)t * B + row) * H + k] + dh_rec[k];
      float o_g = act[k + 3 * H];
      float tc = tanhf(c_new[k]);
      float dc = dht * o_g * (1.f - tc * tc) + dc_rec[k];
      float xhat = (craw[k] - cmean) * crstd;
      // stash dc and xhat in dh_rec/dc_rec temporarily? need them next pass;
      // reuse dgate[0..H) and dgate[H..2H) as scratch (overwritten later)
      dgate[k] = dc;
      dgate[k + H] = xhat;
      m1_loc += lnc_w[k] * dc;
      m2_loc += lnc_w[k] * dc * xhat;
      // accumulate LN_c param grads
      dwc_loc += dc * xhat;   // per-thread, per its k columns
      dbc_loc += dc;
    }
    float m1 = block_sum(m1_loc, scratch) / H;
    float m2 = block_sum(m2_loc, scratch) / H;
    for (int k = tid; k < H; k += NT) {
      float dc = dgate[k];
      float xhat = dgate[k + H];
      float dcraw = (lnc_w[k] * dc - m1 - xhat * m2) * crstd;
      float i_g = act[k], f_g = act[k + H], g_g = act[k + 2 * H];
      float o_g = act[k + 3 * H];
      float dht = dout[((long)t * B + row) * H + k] + dh_rec[k];
      float tc = tanhf(c_new[k]);
      float d_o = dht * tc;
      float d_i = dcraw * g_g;
      float d_f = dcraw * c_prev[k];
      float d_g = dcraw * i_g;
      dc_rec[k] = dcraw * f_g;          // for step t-1
      // pre-activation grads -> dgate buffer (by gate column)
      dgate[k] = d_i * i_g * (1.f - i_g);
      dgate[k + H] = d_f * f_g * (1.f - f_g);
      dgate[k + 2 * H] = d_g * (1.f - g_g * g_g);
      dgate[k + 3 * H] = d_o * o_g * (1.f - o_g);
    }
    __syncthreads();

    // LN_h backward over G
    m1_loc = 0.f; m2_loc = 0.f;
    for (int j = tid; j < G; j += NT) {
      float xhat = (hg[j] - hmean) * hrstd;
      m1_loc += lnh_w[j] * dgate[j];
      m2_loc += lnh_w[j] * dgate[j] * xhat;
    }
    m1 = block_sum(m1_loc, scratch) / G;
    m2 = block_sum(m2_loc, scratch) / G;
    int nown = 0;
    for (int j = tid; j < G; j += NT) {
      float xhat = (hg[j] - hmean) * hrstd;
      float dg_j = dgate[j];
      float v = (lnh_w[j] * dg_j - m1 - xhat * m2) * hrstd;
      dhg[j] = v;
      dhgates_raw[((long)t * B + row) * G + j] = v;
      digates[((long)t * B + row) * G + j] = dg_j;
      dwh_loc[nown] += dg_j * xhat;
      dbh_loc[nown] += dg_j;
      ++nown;
    }
    __syncthreads();

    // dh_prev = dhg @ W_hh  (via W_hh^T rows; 16-byte weight loads)
    for (int k = tid; k < H; k += NT) {
      const uint4* wtrow = reinterpret_cast<const uint4*>(w_hh_t + (long)k * G);
      float acc = 0.f;
      for (int j8 = 0; j8 < G / 8; ++j8) {
        uint4 wbits = wtrow[j8];
        const bf16x2* pw = reinterpret_cast<const bf16x2*>(&wbits);
        const float* dv = dhg + j8 * 8;
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          float2 fw = __bfloat1622float2(pw[i]);
          acc = fmaf(dv[2 * i], fw.x, acc);
          acc = fmaf(dv[2 * i + 1], fw.y, acc);
        }
      }
      dh_rec[k] = acc;
    }
    __syncthreads();
  }

  for (int k = tid; k < H; k += NT) {
    dh0[row * H + k] = dh_rec[k];
    dc0[row * H + k] = dc_rec[k];
  }
  // LN param grads: one atomicAdd per owned column
  int nown = 0;
  for (int j = tid; j < G; j += NT) {
    atomicAdd(&dlnh_w[j], dwh_loc[nown]);
    atomicAdd(&dlnh_b[j], dbh_loc[nown]);
    ++nown;
  }
  for (int k = tid; k < H; k += NT) {
    atomicAdd(&dlnc_w[k], dwc_loc);
    atomicAdd(&dlnc_b[k], dbc_loc);
  }
}


// ---------------------------------------------------------------------------
// Cooperative-launch forward for SMALL BATCH (the learner core: B=16..64).
// The per-row kernel above runs only B workgroups — at B=32 that is 32 of
// 256 CUs.  This variant spreads ONE timestep's h @ W_hh^T across the whole
// chip as an MFMA matmul (M=B, N=4H, K=H), with grid-wide syncs between the
// matmul / gate / cell phases (3 per step) and double-buffered LN stats.
// Outputs are identical to lnlstm_forward_kernel, so the existing backward
// consumes them unchanged.
#include <hip/hip_cooperative_groups.h>

typedef __bf16 lbf16x8 __attribute__((ext_vector_type(8)));
typedef float lf32x4 __attribute__((ext_vector_type(4)));

extern "C" __global__ void lnlstm_forward_coop_kernel(
    const float* __restrict__ igates,     // (T, B, 4H) post-LN_i
    const float* __restrict__ h0,
    const float* __restrict__ c0,
    const bf16* __restrict__ w_hh,        // (4H, H) bf16 row-major
    const float* __restrict__ lnh_w, const float* __restrict__ lnh_b,
    const float* __restrict__ lnc_w, const float* __restrict__ lnc_b,
    float* __restrict__ h_all,            // (T+1, B, H)
    float* __restrict__ c_all,
    float* __restrict__ hgates_raw,       // (T, B, 4H)
    float* __restrict__ cellraw,          // (T, B, H)
    bf16* __restrict__ h_bf,              // ws (Mt*16, H) bf16, h0-filled
    float* __restrict__ o_ws,             // ws (B, H)
    float* __restrict__ hstats,           // ws (2, B, 2) zeroed
    float* __restrict__ cstats,           // ws (2, B, 2) zeroed
    int T, int B, int H) {
  namespace cg = cooperative_groups;
  cg::grid_group grid = cg::this_grid();
  const int G = 4 * H;
  const int Mt = (B + 15) / 16;
  const int Nt = G / 16;
  const int lane = threadIdx.x & 63;
  const int l16 = lane & 15;
  const int lq = lane >> 4;
  const int slot = (int)(blockIdx.x * 4 + (threadIdx.x >> 6));   // wave slot
  const int mt = slot % Mt;
  const int nt = slot / Mt;               // may exceed Nt for pad waves
  const long total_bh = (long)B * H;
  const long tid_g = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long nthreads = (long)gridDim.x * blockDim.x;

  for (int t = 0; t < T; ++t) {
    const int sl = t & 1;
    // ---- P1: hg_raw = h_bf @ W_hh^T (MFMA) + LN_h partials
    if (nt < Nt) {
      lf32x4 acc = (lf32x4){0, 0, 0, 0};
      for (int ks = 0; ks < H / 32; ++ks) {
        lbf16x8 a, b;
        __builtin_memcpy(&a, h_bf + (long)(mt * 16 + l16) * H + ks * 32
                         + lq * 8, 16);
        __builtin_memcpy(&b, w_hh + (long)(nt * 16 + l16) * H + ks * 32
                         + lq * 8, 16);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
      }
      // acc: col = gate (nt*16 + l16), row = b (mt*16 + lq*4 + r)
      for (int r = 0; r < 4; ++r) {
        int b_i = mt * 16 + lq * 4 + r;
        int g_i = nt * 16 + l16;
        if (b_i < B) {
          float v = acc[r];
          hgates_raw[((long)t * B + b_i) * G + g_i] = v;
          // per-(wave, b) partial over this wave's 16 gates
          float s = v, sq = v * v;
          for (int o = 1; o < 16; o <<= 1) {
            s += __shfl_xor(s, o, 64);
            sq += __shfl_xor(sq, o, 64);
          }
          if (l16 == 0) {
            atomicAdd(&hstats[(sl * B + b_i) * 2 + 0], s);
            atomicAdd(&hstats[(sl * B + b_i) * 2 + 1], sq);
          }
        }
      }
    }
    grid.sync();
    // ---- P2: gates + cell-raw + LN_c partials; zero next hstats slot
    for (long i = tid_g; i < total_bh; i += nthreads) {
      int b_i = i / H, h_i = i - (long)(i / H) * H;
      float mean = hstats[(sl * B + b_i) * 2 + 0] / G;
      float var = hstats[(sl * B + b_i) * 2 + 1] / G - mean * mean;
      float rstd = rsqrtf(var + LN_EPS);
      const float* hg = hgates_raw + ((long)t * B + b_i) * G;
      const float* ig = igates + ((long)t * B + b_i) * G;
      float gate[4];
      for (int k = 0; k < 4; ++k) {
        int g_i = k * H + h_i;
        gate[k] = ig[g_i] + (hg[g_i] - mean) * rstd * lnh_w[g_i] + lnh_b[g_i];
      }
      float iv = 1.f / (1.f + __expf(-gate[0]));
      float fv = 1.f / (1.f + __expf(-gate[1]));
      float gv = tanhf(gate[2]);
      float ov = 1.f / (1.f + __expf(-gate[3]));
      float cprev = c_all[((long)t * B + b_i) * H + h_i];
      float craw = fv * cprev + iv * gv;
      cellraw[((long)t * B + b_i) * H + h_i] = craw;
      o_ws[i] = ov;
      float s = craw, sq = craw * craw;
      atomicAdd(&cstats[(sl * B + b_i) * 2 + 0], s);
      atomicAdd(&cstats[(sl * B + b_i) * 2 + 1], sq);
      // zero the other hstats slot for t+1 (its last reader was P2(t-1))
      if (h_i == 0) {
        hstats[((1 - sl) * B + b_i) * 2 + 0] = 0.f;
        hstats[((1 - sl) * B + b_i) * 2 + 1] = 0.f;
      }
    }
    grid.sync();
    // ---- P3: LN_c + h; zero next cstats slot
    for (long i = tid_g; i < total_bh; i += nthreads) {
      int b_i = i / H, h_i = i - (long)(i / H) * H;
      float cmean = cstats[(sl * B + b_i) * 2 + 0] / H;
      float cvar = cstats[(sl * B + b_i) * 2 + 1] / H - cmean * cmean;
      float crstd = rsqrtf(cvar + LN_EPS);
      float craw = cellraw[((long)t * B + b_i) * H + h_i];
      float c_new = (craw - cmean) * crstd * lnc_w[h_i] + lnc_b[h_i];
      float h_new = o_ws[i] * tanhf(c_new);
      c_all[((long)(t + 1) * B + b_i) * H + h_i] = c_new;
      h_all[((long)(t + 1) * B + b_i) * H + h_i] = h_new;
      h_bf[(long)b_i * H + h_i] = __float2bfloat16(h_new);
      if (h_i == 0) {
        cstats[((1 - sl) * B + b_i) * 2 + 0] = 0.f;
        cstats[((1 - sl) * B + b_i) * 2 + 1] = 0.f;
      }
    }
    grid.sync();
  }
}
