// K1: entity-transformer attention, hand-written CDNA4 MFMA kernels.
//
// Replaces F.scaled_dot_product_attention on the entity transformer's hot
// path (reference distar/agent/default/model/module_utils.py:71-151):
// 2 heads x head_dim 128 over <=512 entities with an additive -1e9
// key-padding mask derived from entity_num — passed HERE as an integer per
// batch row, so no (B,1,N,N) mask tensor is ever materialized (the eager
// path allocates ~1 GB of bf16 mask per layer at the SL bench shapes).
//
// Design (flash-style, one pass over K/V tiles with online softmax):
//  * mfma_f32_16x16x32_bf16 everywhere; fp32 accumulation + row stats.
//  * Q,K staged row-major in LDS — for S = Q.K^T BOTH fragments read
//    contiguous 16B (the B-operand wants K[key][d], which row-major K
//    already is).  V is staged transposed (Vt[d][key]) so the PV B-operand
//    reads contiguously too.  P makes a per-wave LDS round-trip to cross
//    from the accumulator layout (col=key) to the A-operand layout
//    (row=q) — each wave touches only its own 16-row band.
//  * XOR swizzle on 16B granules (g ^= row&7) on every LDS tile: a
//    row-major [**][128] bf16 tile read 16-rows-at-a-time is otherwise a
//    16-32 way bank conflict (guide §6 G4).
//  * Masking is soft (-1e9 added pre-softmax), matching the reference's
//    finite-logit convention bit-for-bit in the P≈0 limit; K/V tiles that
//    are fully masked are skipped entirely (exp(-1e9-m) == 0 in fp32, so
//    skipping is exact) — at 256-entity RL batches this halves the work.
//  * Backward: two kernels (dKdV per kv-tile; dQ per q-tile) that
//    recompute P from the saved per-row logsumexp, plus a tiny
//    drow = rowsum(dO*O) pre-pass.
//
// Block = 4 waves (256 threads); each block owns 64 q rows (fwd/dQ) or 64
// kv rows (dKdV) of one (batch, head); waves own 16-row bands.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define D_DIM 128            // head dim (fixed: entity transformer)
#define TILE 64              // q-rows / kv-rows per block
#define NWAVE 4
#define GR 16                // bf16 per 16B granule: 8; granules per row: D/8
#define ROW_GRAN (D_DIM / 8) // 16 granules per 128-elem row
#define NEG_MASK -1e9f

// LDS tile helpers: row-major [rows][128] bf16 with granule XOR swizzle.
__device__ __forceinline__ int swz_off(int row, int col) {
  // byte offset of element (row, col) in a [**][128] bf16 tile
  int g = col >> 3;                 // 16B granule within the row
  g ^= (row & 7);
  return row * 256 + g * 16 + (col & 7) * 2;
}
// [**][64] bf16 tile (8 granules per row)
__device__ __forceinline__ int swz_off64(int row, int col) {
  int g = col >> 3;
  g = (g ^ (row & 7)) & 7;
  return row * 128 + g * 16 + (col & 7) * 2;
}

__device__ __forceinline__ bf16x8 lds_read8(const char* base, int byte_off) {
  return *(const bf16x8*)(base + byte_off);
}

// 8 elements T[row0+j][col] of a row-major [..][128] swizzled tile —
// the transposed-operand fragment (strided u16 reads; saves staging a
// second, transposed copy of the tile and 32 KB of LDS per block).
__device__ __forceinline__ bf16x8 lds_read_col8(const char* base, int row0,
                                                int col) {
  bf16x8 out;
#pragma unroll
  for (int j = 0; j < 8; ++j)
    out[j] = *(const __hip_bfloat16*)(base + swz_off(row0 + j, col));
  return out;
}

// stage a [rows][128] bf16 tile from global (row stride `gstride` elems,
// element base `gbase`) into swizzled LDS; rows >= limit are zero-filled.
__device__ __forceinline__ void stage_tile128(
    char* lds, const __hip_bfloat16* gbase, long gstride, int row0,
    int rows, int limit) {
  // rows*16 granules, 256 threads
  const int tid = threadIdx.x;
  for (int i = tid; i < rows * ROW_GRAN; i += 256) {
    int r = i / ROW_GRAN, g = i % ROW_GRAN;
    int gs = (g ^ (r & 7));
    uint4 val = {0, 0, 0, 0};
    if (row0 + r < limit)
      val = *(const uint4*)(gbase + (long)(row0 + r) * gstride + g * 8);
    *(uint4*)(lds + r * 256 + gs * 16) = val;
  }
}

// stage a transposed tile: global [rows][128] -> LDS [128][rows=64] bf16
// (swizzled [d][key] layout); key rows >= limit zero-filled.
__device__ __forceinline__ void stage_tile_t(
    char* lds, const __hip_bfloat16* gbase, long gstride, int row0,
    int limit) {
  const int tid = threadIdx.x;
  for (int i = tid; i < TILE * ROW_GRAN; i += 256) {
    int key = i / ROW_GRAN, g = i % ROW_GRAN;
    uint4 val = {0, 0, 0, 0};
    if (row0 + key < limit)
      val = *(const uint4*)(gbase + (long)(row0 + key) * gstride + g * 8);
    const __hip_bfloat16* v = (const __hip_bfloat16*)&val;
    for (int j = 0; j < 8; ++j) {
      int d = g * 8 + j;
      *(__hip_bfloat16*)(lds + swz_off64(d, key)) = v[j];
    }
  }
}

extern "C" __global__ __launch_bounds__(256, 2)
void entity_attn_fwd_kernel(
    const __hip_bfloat16* __restrict__ q,   // (B, N, qkv_stride) packed
    const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v,
    const int* __restrict__ entity_num,     // (B,) or nullptr
    __hip_bfloat16* __restrict__ out,       // (B, N, H*D)
    float* __restrict__ lse,                // (B, H, N)
    float scale, int B, int H, int N,
    long b_stride, long n_stride, long h_off_stride,  // elems
    long ob_stride, long on_stride) {
  const int qtile = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int q0 = qtile * TILE;
  if (q0 >= N) return;

  extern __shared__ char lds[];
  char* Qs = lds;                       // [64][128] bf16 swz: 16 KB
  char* Ks = Qs + TILE * 256;           // [64][128] bf16 swz: 16 KB
  char* Vt = Ks + TILE * 256;           // [128][64] bf16 swz: 16 KB
  char* Ps = Vt + D_DIM * 128;          // [64][64]  bf16 swz: 8 KB

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int l16 = lane & 15;
  const int lq = lane >> 4;             // 0..3 quarter-wave
  const int band = wave * 16;

  const int n_eff_raw = entity_num ? entity_num[b] : N;
  const int n_eff = min(max(n_eff_raw, 1), N);   // soft-mask floor: 1 key
  const int n_tiles = (n_eff + TILE - 1) / TILE;

  const __hip_bfloat16* qg = q + (long)b * b_stride + (long)h * h_off_stride;
  const __hip_bfloat16* kg = k + (long)b * b_stride + (long)h * h_off_stride;
  const __hip_bfloat16* vg = v + (long)b * b_stride + (long)h * h_off_stride;

  stage_tile128(Qs, qg, n_stride, q0, TILE, N);

  float m_row[4], l_row[4];
  f32x4 acc_o[8];
  for (int r = 0; r < 4; ++r) { m_row[r] = -INFINITY; l_row[r] = 0.f; }
  for (int t = 0; t < 8; ++t) acc_o[t] = (f32x4){0, 0, 0, 0};

  // the wave's Q fragments are loop-invariant: load once into registers
  bf16x8 qfrag[4];
  __syncthreads();
  for (int ks = 0; ks < 4; ++ks)
    qfrag[ks] = lds_read8(Qs, swz_off(band + l16, ks * 32 + lq * 8));

  for (int kt = 0; kt < n_tiles; ++kt) {
    const int kv0 = kt * TILE;
    __syncthreads();
    stage_tile128(Ks, kg, n_stride, kv0, TILE, N);
    stage_tile_t(Vt, vg, n_stride, kv0, N);
    __syncthreads();

    // S band: 16 q-rows x 64 keys = 4 n-tiles, k over 128 dims = 4 steps
    f32x4 acc_s[4];
    for (int nt = 0; nt < 4; ++nt) acc_s[nt] = (f32x4){0, 0, 0, 0};
    for (int ks = 0; ks < 4; ++ks) {
      bf16x8 a = qfrag[ks];
      for (int nt = 0; nt < 4; ++nt) {
        bf16x8 bfr = lds_read8(Ks, swz_off(nt * 16 + l16, ks * 32 + lq * 8));
        acc_s[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfr,
                                                            acc_s[nt], 0, 0, 0);
      }
    }
    // scale + mask; rows of the acc: q = band + lq*4 + r; col key = nt*16+l16
    float p[4][4];           // [nt][r]
    float tile_max[4];       // per r
    for (int r = 0; r < 4; ++r) tile_max[r] = -INFINITY;
    for (int nt = 0; nt < 4; ++nt) {
      int key_g = kv0 + nt * 16 + l16;
      float mask_add = (key_g < n_eff) ? 0.f : NEG_MASK;
      for (int r = 0; r < 4; ++r) {
        float s = acc_s[nt][r] * scale + mask_add;
        p[nt][r] = s;
        tile_max[r] = fmaxf(tile_max[r], s);
      }
    }
    for (int r = 0; r < 4; ++r) {
      for (int off = 1; off < 16; off <<= 1)
        tile_max[r] = fmaxf(tile_max[r], __shfl_xor(tile_max[r], off, 64));
      float m_new = fmaxf(m_row[r], tile_max[r]);
      float corr = __expf(m_row[r] - m_new);     // exp(-inf)=0 on first tile
      float rsum = 0.f;
      for (int nt = 0; nt < 4; ++nt) {
        float e = __expf(p[nt][r] - m_new);
        p[nt][r] = e;
        rsum += e;
      }
      for (int off = 1; off < 16; off <<= 1)
        rsum += __shfl_xor(rsum, off, 64);
      l_row[r] = l_row[r] * corr + rsum;
      m_row[r] = m_new;
      for (int t = 0; t < 8; ++t) acc_o[t][r] *= corr;
    }
    // P band -> LDS (own band only; bf16)
    for (int nt = 0; nt < 4; ++nt)
      for (int r = 0; r < 4; ++r)
        *(__hip_bfloat16*)(Ps + swz_off64(band + lq * 4 + r, nt * 16 + l16)) =
            __float2bfloat16(p[nt][r]);
    __builtin_amdgcn_s_waitcnt(0);     // drain own ds_writes before reads
    // PV: O band (16 x 128) += P(16x64) . V(64x128)
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 a = lds_read8(Ps, swz_off64(band + l16, ks * 32 + lq * 8));
      for (int t = 0; t < 8; ++t) {
        bf16x8 bfr = lds_read8(Vt, swz_off64(t * 16 + l16, ks * 32 + lq * 8));
        acc_o[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfr,
                                                           acc_o[t], 0, 0, 0);
      }
    }
  }

  // epilogue: O /= l, write out + lse
  for (int r = 0; r < 4; ++r) {
    int qg_row = q0 + band + lq * 4 + r;
    if (qg_row >= N) continue;
    float inv_l = (l_row[r] > 0.f) ? 1.f / l_row[r] : 0.f;
    __hip_bfloat16* orow = out + (long)b * ob_stride +
        (long)qg_row * on_stride + (long)h * D_DIM;
    for (int t = 0; t < 8; ++t)
      orow[t * 16 + l16] = __float2bfloat16(acc_o[t][r] * inv_l);
    if (l16 == 0)
      lse[((long)b * H + h) * N + qg_row] = m_row[r] + __logf(l_row[r]);
  }
}

// drow[b,h,n] = sum_d dO[b,n,h*D+d] * O[b,n,h*D+d]  (fp32)
extern "C" __global__ void attn_drow_kernel(
    const __hip_bfloat16* __restrict__ dout,
    const __hip_bfloat16* __restrict__ out,
    float* __restrict__ drow,
    int B, int H, int N, long ob_stride, long on_stride) {
  long row = blockIdx.x * 4 + (threadIdx.x >> 6);     // one wave per row
  long total = (long)B * H * N;
  int lane = threadIdx.x & 63;
  for (; row < total; row += (long)gridDim.x * 4) {
    int n = row % N;
    int h = (row / N) % H;
    long b = row / ((long)N * H);
    const __hip_bfloat16* dp = dout + b * ob_stride + (long)n * on_stride
        + (long)h * D_DIM;
    const __hip_bfloat16* op = out + b * ob_stride + (long)n * on_stride
        + (long)h * D_DIM;
    float acc = __bfloat162float(dp[lane]) * __bfloat162float(op[lane]) +
                __bfloat162float(dp[lane + 64]) * __bfloat162float(op[lane + 64]);
    for (int off = 1; off < 64; off <<= 1)
      acc += __shfl_xor(acc, off, 64);
    if (lane == 0) drow[row] = acc;
  }
}

// dK/dV: one block per (b, h, kv-tile); loops q-tiles.
extern "C" __global__ __launch_bounds__(256, 2)
void entity_attn_bwd_kv_kernel(
    const __hip_bfloat16* __restrict__ q,
    const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v,
    const __hip_bfloat16* __restrict__ dout,   // (B, N, H*D)
    const float* __restrict__ lse,             // (B, H, N)
    const float* __restrict__ drow,            // (B, H, N)
    const int* __restrict__ entity_num,
    __hip_bfloat16* __restrict__ dk,           // packed dqkv k-slot
    __hip_bfloat16* __restrict__ dv,           // packed dqkv v-slot
    float scale, int B, int H, int N,
    long b_stride, long n_stride, long h_off_stride,
    long ob_stride, long on_stride,            // dout/out layout
    long gb_stride, long gn_stride) {          // dqkv layout
  const int kvt = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int kv0 = kvt * TILE;
  if (kv0 >= N) return;

  const int n_eff_raw = entity_num ? entity_num[b] : N;
  const int n_eff = min(max(n_eff_raw, 1), N);
  // fully-masked kv tile: P column is exactly 0 -> dK = dV = 0
  extern __shared__ char lds[];
  char* Ks  = lds;                      // [64][128] 16 KB
  char* Vs  = Ks + TILE * 256;          // [64][128] 16 KB
  char* Qs  = Vs + TILE * 256;          // [64][128] 16 KB (per q-tile)
  char* dOs = Qs + TILE * 256;          // [64][128] 16 KB
  char* Ps  = dOs + TILE * 256;         // [64][64]   8 KB (P', then dS')
  float* lse_s  = (float*)(Ps + TILE * 128);   // [64]
  float* drow_s = lse_s + TILE;                // [64]

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int l16 = lane & 15;
  const int lq = lane >> 4;
  const int band = wave * 16;           // kv band of this wave
  const int tid = threadIdx.x;

  const __hip_bfloat16* qg = q + (long)b * b_stride + (long)h * h_off_stride;
  const __hip_bfloat16* kg = k + (long)b * b_stride + (long)h * h_off_stride;
  const __hip_bfloat16* vg = v + (long)b * b_stride + (long)h * h_off_stride;
  const __hip_bfloat16* dog = dout + (long)b * ob_stride + (long)h * D_DIM;
  const float* lse_g = lse + ((long)b * H + h) * N;
  const float* drow_g = drow + ((long)b * H + h) * N;

  const bool dead = kv0 >= n_eff;       // whole tile masked

  stage_tile128(Ks, kg, n_stride, kv0, TILE, N);
  stage_tile128(Vs, vg, n_stride, kv0, TILE, N);   // V row-major (dP' B-op)
  // (K/V fragment hoisting here spills — 16 extra VGPRs tip this kernel
  // past 256 with its two 8-tile accumulators; LDS re-reads are cheaper)

  f32x4 acc_dvT[8], acc_dk[8];
  for (int t = 0; t < 8; ++t) {
    acc_dvT[t] = (f32x4){0, 0, 0, 0};
    acc_dk[t] = (f32x4){0, 0, 0, 0};
  }

  const int q_tiles = (N + TILE - 1) / TILE;
  for (int qt = 0; qt < q_tiles && !dead; ++qt) {
    const int q0 = qt * TILE;
    __syncthreads();
    stage_tile128(Qs, qg, n_stride, q0, TILE, N);
    stage_tile128(dOs, dog, on_stride, q0, TILE, N);
    for (int i = tid; i < TILE; i += 256) {
      lse_s[i] = (q0 + i < N) ? lse_g[q0 + i] : 0.f;
      drow_s[i] = (q0 + i < N) ? drow_g[q0 + i] : 0.f;
    }
    __syncthreads();

    // S' = K.Q^T band: m=key (band), n=q (4 tiles), k=d (4 steps)
    f32x4 acc_s[4];
    for (int nt = 0; nt < 4; ++nt) acc_s[nt] = (f32x4){0, 0, 0, 0};
    for (int ks = 0; ks < 4; ++ks) {
      bf16x8 a = lds_read8(Ks, swz_off(band + l16, ks * 32 + lq * 8));
      for (int nt = 0; nt < 4; ++nt) {
        bf16x8 bq = lds_read8(Qs, swz_off(nt * 16 + l16, ks * 32 + lq * 8));
        acc_s[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bq,
                                                            acc_s[nt], 0, 0, 0);
      }
    }
    // P' = exp(s*scale + mask - lse[q]); acc rows = key band+lq*4+r, col=q
    float pprime[4][4];
    for (int nt = 0; nt < 4; ++nt) {
      int qcol = q0 + nt * 16 + l16;
      float l = lse_s[nt * 16 + l16];
      bool qvalid = qcol < N;
      for (int r = 0; r < 4; ++r) {
        int key_g = kv0 + band + lq * 4 + r;
        float mask_add = (key_g < n_eff) ? 0.f : NEG_MASK;
        float e = qvalid ? __expf(acc_s[nt][r] * scale + mask_add - l) : 0.f;
        pprime[nt][r] = e;
      }
    }
    // write P' band -> Ps[key][q]
    for (int nt = 0; nt < 4; ++nt)
      for (int r = 0; r < 4; ++r)
        *(__hip_bfloat16*)(Ps + swz_off64(band + lq * 4 + r, nt * 16 + l16)) =
            __float2bfloat16(pprime[nt][r]);
    __builtin_amdgcn_s_waitcnt(0);
    // dV^T += dO^T . P'^T : m=d (8 tiles), n=key(16 of this band), k=q
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 bp = lds_read8(Ps, swz_off64(band + l16, ks * 32 + lq * 8));
      for (int t = 0; t < 8; ++t) {
        bf16x8 a = lds_read_col8(dOs, ks * 32 + lq * 8, t * 16 + l16);
        acc_dvT[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bp,
                                                             acc_dvT[t], 0, 0, 0);
      }
    }
    // dP' = V . dO^T : m=key band, n=q 4 tiles, k=d 4 steps
    f32x4 acc_dp[4];
    for (int nt = 0; nt < 4; ++nt) acc_dp[nt] = (f32x4){0, 0, 0, 0};
    for (int ks = 0; ks < 4; ++ks) {
      bf16x8 a = lds_read8(Vs, swz_off(band + l16, ks * 32 + lq * 8));
      for (int nt = 0; nt < 4; ++nt) {
        bf16x8 bdo = lds_read8(dOs, swz_off(nt * 16 + l16, ks * 32 + lq * 8));
        acc_dp[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bdo,
                                                             acc_dp[nt], 0, 0, 0);
      }
    }
    // dS' = P' * (dP' - drow[q]) * scale  -> Ps (reuse, own band)
    __syncthreads();      // all waves done reading P' (dV step) first
    for (int nt = 0; nt < 4; ++nt) {
      float dr = drow_s[nt * 16 + l16];
      for (int r = 0; r < 4; ++r) {
        float ds = pprime[nt][r] * (acc_dp[nt][r] - dr) * scale;
        *(__hip_bfloat16*)(Ps + swz_off64(band + lq * 4 + r, nt * 16 + l16)) =
            __float2bfloat16(ds);
      }
    }
    __builtin_amdgcn_s_waitcnt(0);
    // dK += dS' . Q : m=key band, n=d 8 tiles, k=q 2 steps (B = Q^T via
    // strided column reads of the row-major Q tile)
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 a = lds_read8(Ps, swz_off64(band + l16, ks * 32 + lq * 8));
      for (int t = 0; t < 8; ++t) {
        bf16x8 bq = lds_read_col8(Qs, ks * 32 + lq * 8, t * 16 + l16);
        acc_dk[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bq,
                                                            acc_dk[t], 0, 0, 0);
      }
    }
  }

  // write dK (acc rows = key, col = d per tile) and dV (acc_dvT rows = d,
  // col = key).  dk/dv are (B, N, H*D) fresh buffers.
  for (int r = 0; r < 4; ++r) {
    // dK: row key = band + lq*4 + r
    int key_g = kv0 + band + lq * 4 + r;
    if (key_g < N) {
      __hip_bfloat16* krow = dk + (long)b * gb_stride +
          (long)key_g * gn_stride + (long)h * D_DIM;
      for (int t = 0; t < 8; ++t)
        krow[t * 16 + l16] = __float2bfloat16(dead ? 0.f : acc_dk[t][r]);
    }
  }
  // dV^T: acc rows = d = t*16 + lq*4 + r, col = key = band? NO: n=key of
  // this band -> col = lane&15 indexes 16 keys of the band.
  for (int t = 0; t < 8; ++t) {
    for (int r = 0; r < 4; ++r) {
      int d = t * 16 + lq * 4 + r;
      int key_g = kv0 + band + l16;
      if (key_g < N) {
        __hip_bfloat16* vrow = dv + (long)b * gb_stride +
            (long)key_g * gn_stride + (long)h * D_DIM;
        vrow[d] = __float2bfloat16(dead ? 0.f : acc_dvT[t][r]);
      }
    }
  }
}

// dQ: one block per (b, h, q-tile); loops kv-tiles.
extern "C" __global__ __launch_bounds__(256, 2)
void entity_attn_bwd_q_kernel(
    const __hip_bfloat16* __restrict__ q,
    const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v,
    const __hip_bfloat16* __restrict__ dout,
    const float* __restrict__ lse,
    const float* __restrict__ drow,
    const int* __restrict__ entity_num,
    __hip_bfloat16* __restrict__ dq,          // packed dqkv q-slot
    float scale, int B, int H, int N,
    long b_stride, long n_stride, long h_off_stride,
    long ob_stride, long on_stride,            // dout/out layout
    long gb_stride, long gn_stride) {          // dqkv layout
  const int qtile = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int q0 = qtile * TILE;
  if (q0 >= N) return;

  extern __shared__ char lds[];
  char* Qs  = lds;                      // [64][128] 16 KB
  char* dOs = Qs + TILE * 256;          // [64][128] 16 KB
  char* Ks  = dOs + TILE * 256;         // [64][128] 16 KB
  char* Vs  = Ks + TILE * 256;          // [64][128] 16 KB
  char* Ps  = Vs + TILE * 256;          // [64][64]   8 KB
  float* lse_s  = (float*)(Ps + TILE * 128);
  float* drow_s = lse_s + TILE;

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int l16 = lane & 15;
  const int lq = lane >> 4;
  const int band = wave * 16;           // q band
  const int tid = threadIdx.x;

  const int n_eff_raw = entity_num ? entity_num[b] : N;
  const int n_eff = min(max(n_eff_raw, 1), N);
  const int n_tiles = (n_eff + TILE - 1) / TILE;

  const __hip_bfloat16* qg = q + (long)b * b_stride + (long)h * h_off_stride;
  const __hip_bfloat16* kg = k + (long)b * b_stride + (long)h * h_off_stride;
  const __hip_bfloat16* vg = v + (long)b * b_stride + (long)h * h_off_stride;
  const __hip_bfloat16* dog = dout + (long)b * ob_stride + (long)h * D_DIM;
  const float* lse_g = lse + ((long)b * H + h) * N;
  const float* drow_g = drow + ((long)b * H + h) * N;

  stage_tile128(Qs, qg, n_stride, q0, TILE, N);
  stage_tile128(dOs, dog, on_stride, q0, TILE, N);
  for (int i = tid; i < TILE; i += 256) {
    lse_s[i] = (q0 + i < N) ? lse_g[q0 + i] : 0.f;
    drow_s[i] = (q0 + i < N) ? drow_g[q0 + i] : 0.f;
  }
  // loop-invariant A-operand fragments (this wave's q band) -> registers
  __syncthreads();
  bf16x8 qfrag[4], dofrag[4];
  for (int ks = 0; ks < 4; ++ks) {
    qfrag[ks] = lds_read8(Qs, swz_off(band + l16, ks * 32 + lq * 8));
    dofrag[ks] = lds_read8(dOs, swz_off(band + l16, ks * 32 + lq * 8));
  }

  f32x4 acc_dq[8];
  for (int t = 0; t < 8; ++t) acc_dq[t] = (f32x4){0, 0, 0, 0};

  for (int kt = 0; kt < n_tiles; ++kt) {
    const int kv0 = kt * TILE;
    __syncthreads();
    stage_tile128(Ks, kg, n_stride, kv0, TILE, N);
    stage_tile128(Vs, vg, n_stride, kv0, TILE, N);
    __syncthreads();

    // S band = Q.K^T : m=q band, n=key 4 tiles, k=d 4 steps
    f32x4 acc_s[4];
    for (int nt = 0; nt < 4; ++nt) acc_s[nt] = (f32x4){0, 0, 0, 0};
    for (int ks = 0; ks < 4; ++ks) {
      bf16x8 a = qfrag[ks];
      for (int nt = 0; nt < 4; ++nt) {
        bf16x8 bk = lds_read8(Ks, swz_off(nt * 16 + l16, ks * 32 + lq * 8));
        acc_s[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bk,
                                                            acc_s[nt], 0, 0, 0);
      }
    }
    // dP band = dO.V^T : same geometry, B from Vs (row-major V)
    f32x4 acc_dp[4];
    for (int nt = 0; nt < 4; ++nt) acc_dp[nt] = (f32x4){0, 0, 0, 0};
    for (int ks = 0; ks < 4; ++ks) {
      bf16x8 a = dofrag[ks];
      for (int nt = 0; nt < 4; ++nt) {
        bf16x8 bv = lds_read8(Vs, swz_off(nt * 16 + l16, ks * 32 + lq * 8));
        acc_dp[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bv,
                                                             acc_dp[nt], 0, 0, 0);
      }
    }
    // dS = P * (dP - drow) * scale; P = exp(s*scale+mask-lse); acc rows=q
    for (int nt = 0; nt < 4; ++nt) {
      int key_g = kv0 + nt * 16 + l16;
      float mask_add = (key_g < n_eff) ? 0.f : NEG_MASK;
      for (int r = 0; r < 4; ++r) {
        int qrow = band + lq * 4 + r;
        bool qvalid = q0 + qrow < N;
        float pv = qvalid
            ? __expf(acc_s[nt][r] * scale + mask_add - lse_s[qrow]) : 0.f;
        float ds = pv * (acc_dp[nt][r] - drow_s[qrow]) * scale;
        *(__hip_bfloat16*)(Ps + swz_off64(qrow, nt * 16 + l16)) =
            __float2bfloat16(ds);
      }
    }
    __builtin_amdgcn_s_waitcnt(0);
    // dQ band += dS . K : m=q band, n=d 8 tiles, k=key 2 steps (B = K^T
    // via strided column reads of the row-major K tile)
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 a = lds_read8(Ps, swz_off64(band + l16, ks * 32 + lq * 8));
      for (int t = 0; t < 8; ++t) {
        bf16x8 bk = lds_read_col8(Ks, ks * 32 + lq * 8, t * 16 + l16);
        acc_dq[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bk,
                                                            acc_dq[t], 0, 0, 0);
      }
    }
  }

  for (int r = 0; r < 4; ++r) {
    int qg_row = q0 + band + lq * 4 + r;
    if (qg_row >= N) continue;
    __hip_bfloat16* qrow_p = dq + (long)b * gb_stride +
        (long)qg_row * gn_stride + (long)h * D_DIM;
    for (int t = 0; t < 8; ++t)
      qrow_p[t * 16 + l16] = __float2bfloat16(acc_dq[t][r]);
  }
}

// layout self-test: one 16x16x32 MFMA, D = A(16x32) . B(32x16), all bf16
// row-major inputs, fp32 out.  Catches any fragment-layout mistake in
// isolation from the attention logic.
extern "C" __global__ void mfma_selftest_kernel(
    const __hip_bfloat16* __restrict__ A,   // (16, 32) row-major
    const __hip_bfloat16* __restrict__ Bm,  // (16, 32) row-major = B^T
    float* __restrict__ Dm) {               // (16, 16): A . Bm^T
  int lane = threadIdx.x & 63;
  int l16 = lane & 15, lq = lane >> 4;
  bf16x8 a = *(const bf16x8*)(A + l16 * 32 + lq * 8);
  bf16x8 b = *(const bf16x8*)(Bm + l16 * 32 + lq * 8);
  f32x4 acc = (f32x4){0, 0, 0, 0};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  for (int r = 0; r < 4; ++r)
    Dm[(lq * 4 + r) * 16 + l16] = acc[r];
}
