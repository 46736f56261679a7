// K4: spatial conv stack — hand-written CDNA4 MFMA implicit-GEMM conv,
// NCHW bf16, stride 1 (the whole DI-star conv surface: 1x1 projections and
// 3x3 pad-1 convs; downsampling is MaxPool, reference
// obs_encoder/spatial_encoder.py:20-90, head/action_arg_head.py:417-450).
//
// Formulation: out[b,co,p] = sum_k im2col[b,p,k] . W[co,k], k = ci*KH*KW +
// offset.  A-operand = LDS-staged im2col tile [64 px][64 k-chunk] (the
// gather does the halo/bounds logic, so NO NHWC transposes and no
// workspace, unlike the MIOpen path it replaces); B-operand = packed
// weights Wp[Cout][Kpad] read straight from global (small, L2-resident;
// the B-fragment wants per-lane contiguous k, which row-major Wp gives).
// Bias + optional ReLU fused into the epilogue.
//
// Backward-data reuses the SAME kernel with flipped/transposed packed
// weights (stride-1 conv duality).  Backward-weight is a second kernel:
// dW[k,co] += im2col^T . dOut — A = LDS im2col-transposed tile, B = dOut
// read from global NCHW (for fixed co, pixels are contiguous), fp32
// atomics into dW.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

typedef __bf16 bf16x8c __attribute__((ext_vector_type(8)));
typedef float f32x4c __attribute__((ext_vector_type(4)));

#define CTILE 64                  // px per block (fwd) / k-rows (wgrad)
#define KC 128                    // k-chunk (4 MFMA k-steps per barrier)

// [64][128] bf16 tile, 256-byte rows, granule XOR swizzle
__device__ __forceinline__ int cswz(int row, int col) {
  int g = (col >> 3) ^ (row & 7);   // 16 granules/row; XOR spreads banks
  return row * 256 + g * 16 + (col & 7) * 2;
}

__device__ __forceinline__ bf16x8c clds8(const char* base, int off) {
  return *(const bf16x8c*)(base + off);
}

// im2col element (p, k) of image (b): in[b][ci][y+dy][x+dx] with
// k = ci*KH*KW + (dy+padH)*KW + (dx+padW); 0 outside / beyond K_real.
__device__ __forceinline__ float im2col_elem(
    const __hip_bfloat16* __restrict__ inb, int p, int k,
    int H, int W, int KH, int KW, int padH, int padW, int K_real) {
  if (k >= K_real) return 0.f;
  int ci = k / (KH * KW);
  int off = k % (KH * KW);
  int y = p / W + off / KW - padH;
  int x = p % W + off % KW - padW;
  if (y < 0 || y >= H || x < 0 || x >= W) return 0.f;
  return __bfloat162float(inb[((long)ci * H + y) * W + x]);
}


// one staged im2col 8-px group for k row `k` of image `inb` (shared by
// the fwd and wgrad staging loops)
__device__ __forceinline__ void im2col_gather8(
    __hip_bfloat16* vals, const __hip_bfloat16* __restrict__ inb,
    int k, int pbase, int H, int W, int HW, int KH, int KW,
    int padH, int padW, int K_real, int kwin, int wrecip) {
  for (int j = 0; j < 8; ++j) vals[j] = __float2bfloat16(0.f);
  if (k >= K_real) return;
  int ci, dy, dx;
  if (kwin == 1) {
    ci = k; dy = -padH; dx = -padW;
  } else {                           // 3x3: mul-shift div by 9 / 3
    ci = (k * 7282) >> 16;
    int off = k - ci * 9;
    dy = ((off * 21846) >> 16) - padH;
    dx = off - ((off * 21846) >> 16) * 3 - padW;
  }
  int y0 = (int)(((long)pbase * wrecip) >> 20);
  int x0 = pbase - y0 * W;
  int y = y0 + dy;
  int x = x0 + dx;
  const __hip_bfloat16* src = inb + ((long)ci * H + y) * W + x;
  if (pbase + 7 < HW && x0 + 7 < W && y >= 0 && y < H &&
      x >= 0 && x + 7 < W) {
    __builtin_memcpy(vals, src, 16);
  } else if (kwin == 1 && pbase + 8 <= HW) {
    // 1x1 im2col is x itself: contiguous across row boundaries
    __builtin_memcpy(vals, inb + (long)ci * HW + pbase, 16);
  } else if (W >= 8) {
    // branchless: <= 1 row crossing per 8-run when W >= 8
    const __hip_bfloat16* cib = inb + (long)ci * H * W;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int wrap = (x0 + j >= W) ? 1 : 0;
      const int yj = y0 + wrap + dy;
      const int xj = x0 + j - wrap * W + dx;
      const bool ok = (pbase + j < HW) && (yj >= 0) && (yj < H) &&
                      (xj >= 0) && (xj < W);
      const __hip_bfloat16 t = cib[ok ? ((long)yj * W + xj) : 0];
      if (ok) vals[j] = t;
    }
  } else {
    const __hip_bfloat16* cib = inb + (long)ci * H * W;
    for (int j = 0; j < 8; ++j) {
      if (pbase + j < HW) {
        int yj = y0 + dy, xj = x0 + dx;
        if (yj >= 0 && yj < H && xj >= 0 && xj < W)
          vals[j] = cib[(long)yj * W + xj];
      }
      if (++x0 == W) { x0 = 0; ++y0; }
    }
  }
}

extern "C" __global__ __launch_bounds__(256, 4)
void conv2d_fwd_kernel(
    const __hip_bfloat16* __restrict__ input,   // (B, Cin, H, W)
    const __hip_bfloat16* __restrict__ wp,      // (Cout, Kpad) packed bf16
    const float* __restrict__ bias,             // (Cout) or nullptr
    __hip_bfloat16* __restrict__ out,           // (B, Cout, H, W)
    int B, int Cin, int Cout, int H, int W,
    int KH, int KW, int padH, int padW, int Kpad, int relu) {
  const int HW = H * W;
  const int p0 = blockIdx.x * CTILE;
  const int n0 = blockIdx.y * 128;      // one block covers up to 128 cout
  const long b = blockIdx.z;
  if (p0 >= HW || n0 >= Cout) return;
  const int K_real = Cin * KH * KW;
  const int wrecip = (1048576 + W - 1) / W;     // 2^20/W, exact for p<2^31/recip

  extern __shared__ char lds[];                 // A tile [64px][KC] 8 KB
  const __hip_bfloat16* inb = input + b * Cin * HW;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int l16 = lane & 15;
  const int lq = lane >> 4;
  const int band = wave * 16;
  const int tid = threadIdx.x;

  // Swapped MFMA: D[m=co][n=px] = W[co,k] . im2col^T[k,px] — the A
  // operand streams straight from packed global weights (L1/L2-resident,
  // per-lane contiguous k) and the accumulator's col=lane&15 becomes the
  // PIXEL, so the epilogue stores 16 consecutive pixels per lane group.
  // Each WAVE owns one 16-pixel band and ALL 64 cout rows (4 m-tiles):
  // one LDS B-fragment feeds 4 MFMAs (the A-operands are cheap global
  // reads), 4:1 MFMA:ds_read instead of 1:1.
  // two 64-cout halves per block: one staged im2col tile feeds 2x the
  // MFMAs (PMC r2n: staging VALU dominated at 59 insts per MFMA)
  f32x4c acc[2][4];
  for (int h = 0; h < 2; ++h)
    for (int nt = 0; nt < 4; ++nt) acc[h][nt] = (f32x4c){0, 0, 0, 0};

  // skip whole chunks/k-rows that are pure padding (e.g. the 56->128 pad
  // of the 1x1 projection would otherwise double the staged volume)
  const int k_hi = min(Kpad, (K_real + 31) / 32 * 32);
  // reciprocal constants: k/(KH*KW) and off/KW via mul+shift (PMC showed
  // the staging's integer divisions + column-major LDS writes made the
  // kernel 69:1 VALU:MFMA with 32-way write conflicts)
  const int kwin = KH * KW;
  for (int k0 = 0; k0 < k_hi; k0 += KC) {
    const int krows = min(k_hi - k0, KC);
    __syncthreads();
    // k-major staging with kk FASTEST across threads: the 8-pixel global
    // run stays one vector load, and a wave's 64 LDS writes span all 32
    // banks (kk>>3 varies per lane; px-major order had 16-way conflicts,
    // px-major single-writes lost the vectorized global reads — PMC r2l/m)
    const int krecip = (65536 + krows - 1) / krows;
    // 2-task batches with the LDS writes hoisted: iteration i's gather
    // must not serialize behind iteration i-1's LDS store (PMC r2jj)
    for (int t0 = tid; t0 < krows * 8; t0 += 512) {
      __hip_bfloat16 vals[2][8];
      int gs[2], kks[2];
      #pragma unroll
      for (int u = 0; u < 2; ++u) {
        const int task = t0 + u * 256;
        gs[u] = -1;
        if (task < krows * 8) {
          const int g = (task * krecip) >> 16;
          const int kk = task - g * krows;
          gs[u] = g; kks[u] = kk;
          im2col_gather8(vals[u], inb, k0 + kk, p0 + g * 8, H, W, HW,
                         KH, KW, padH, padW, K_real, kwin, wrecip);
        }
      }
      #pragma unroll
      for (int u = 0; u < 2; ++u) {
        if (gs[u] < 0) continue;
        for (int j = 0; j < 8; ++j)
          *(__hip_bfloat16*)(lds + cswz(gs[u] * 8 + j, kks[u])) = vals[u][j];
      }
    }
    __syncthreads();
    const int ks_count = min(k_hi - k0, KC) / 32;
    for (int ks = 0; ks < ks_count; ++ks) {
      bf16x8c bi[4];
      for (int nt = 0; nt < 4; ++nt)
        bi[nt] = clds8(lds, cswz(nt * 16 + l16, ks * 32 + lq * 8));
      for (int h = 0; h < 2; ++h) {
        // waves/halves whose whole co band is beyond Cout skip compute
        if (n0 + h * 64 + band >= Cout) continue;
        int co_a = n0 + h * 64 + band + l16;
        bf16x8c a = (co_a < Cout)
            ? *(const bf16x8c*)(wp + (long)co_a * Kpad + k0 + ks * 32 + lq * 8)
            : (bf16x8c){0, 0, 0, 0, 0, 0, 0, 0};
        for (int nt = 0; nt < 4; ++nt)
          acc[h][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a, bi[nt], acc[h][nt], 0, 0, 0);
      }
    }
  }
  // epilogue: acc rows = co (band + lq*4 + r), cols = px (nt*16 + l16):
  // 16-lane groups store 32 contiguous bytes
  for (int h = 0; h < 2; ++h) {
    for (int r = 0; r < 4; ++r) {
      int co = n0 + h * 64 + band + lq * 4 + r;
      if (co >= Cout) continue;
      float bv = bias ? bias[co] : 0.f;
      for (int nt = 0; nt < 4; ++nt) {
        int p = p0 + nt * 16 + l16;
        if (p >= HW) continue;
        float v = acc[h][nt][r] + bv;
        if (relu) v = fmaxf(v, 0.f);
        out[(b * Cout + co) * HW + p] = __float2bfloat16(v);
      }
    }
  }
}

// dW[k][co] += sum_px im2col[px][k] * dout[co][px]   (fp32 atomics)
extern "C" __global__ __launch_bounds__(256, 2)
void conv2d_wgrad_kernel(
    const __hip_bfloat16* __restrict__ input,   // (B, Cin, H, W)
    const __hip_bfloat16* __restrict__ dout,    // (B, Cout, H, W)
    float* __restrict__ dwp,                    // (Kpad, Cout) fp32
    float* __restrict__ dbias,                  // (Cout) fp32 or nullptr
    int B, int Cin, int Cout, int H, int W,
    int KH, int KW, int padH, int padW, int Kpad, int ipb) {
  const int HW = H * W;
  const int k_base = blockIdx.x * CTILE;        // k tile (rows of dW)
  const int b_base = blockIdx.z * ipb;          // image range
  const int K_real = Cin * KH * KW;
  if (k_base >= K_real) return;                 // pure-padding k tile
  const int n_tiles_co = (Cout + 15) / 16;      // <= 8 (Cout <= 128)
  const int wrecip = (1048576 + W - 1) / W;

  extern __shared__ char lds[];                 // A^T tile [64 k][64 px] 8 KB
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int l16 = lane & 15;
  const int lq = lane >> 4;
  const int band = wave * 16;                   // k band of this wave
  const int tid = threadIdx.x;

  f32x4c acc[8];                        // co tiles (<= 128 cout)
  for (int nt = 0; nt < 8; ++nt) acc[nt] = (f32x4c){0, 0, 0, 0};
  // bias grad rides along in the k-tile-0 blocks (wave 0's B fragments
  // cover each dout element of the image range exactly once) — replaces
  // a separate full-tensor reduce per conv
  const bool do_db = dbias && blockIdx.x == 0 && wave == 0;
  float acc_db[8] = {};

  for (int bi = 0; bi < ipb && b_base + bi < B; ++bi) {
    const long b = b_base + bi;
    const __hip_bfloat16* inb = input + b * Cin * HW;
    const __hip_bfloat16* dob = dout + b * Cout * HW;
    for (int p0 = 0; p0 < HW; p0 += KC) {
      __syncthreads();
      // stage im2col^T chunk (2-task batches, writes hoisted — see fwd)
      for (int t0 = tid; t0 < CTILE * (KC / 8); t0 += 512) {
        __hip_bfloat16 vals[2][8];
        int kks[2], gg[2];
        #pragma unroll
        for (int u = 0; u < 2; ++u) {
          const int task = t0 + u * 256;
          gg[u] = -1;
          if (task < CTILE * (KC / 8)) {
            const int kk = task >> 4, g = task & 15;
            kks[u] = kk; gg[u] = g;
            im2col_gather8(vals[u], inb, k_base + kk, p0 + g * 8, H, W, HW,
                           KH, KW, padH, padW, K_real, KH * KW, wrecip);
          }
        }
        #pragma unroll
        for (int u = 0; u < 2; ++u)
          if (gg[u] >= 0)
            __builtin_memcpy(lds + cswz(kks[u], gg[u] * 8), vals[u], 16);
      }
      __syncthreads();
      for (int ks = 0; ks < 4; ++ks) {
        bf16x8c a = clds8(lds, cswz(band + l16, ks * 32 + lq * 8));
        for (int nt = 0; nt < n_tiles_co; ++nt) {
          int co = nt * 16 + l16;
          bf16x8c bdo = (bf16x8c){0, 0, 0, 0, 0, 0, 0, 0};
          if (co < Cout) {
            int p = p0 + ks * 32 + lq * 8;
            if (p + 8 <= HW) {
              // HW*2B is not always 16B-aligned per-row: memcpy lets the
              // compiler emit the widest legal loads
              __builtin_memcpy(&bdo, dob + (long)co * HW + p, 16);
            } else {
              for (int j = 0; j < 8; ++j)
                if (p + j < HW)
                  bdo[j] = ((const __bf16*)dob)[(long)co * HW + p + j];
            }
          }
          acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bdo, acc[nt],
                                                            0, 0, 0);
          if (do_db) {
            float sdb = 0.f;
            for (int j = 0; j < 8; ++j) sdb += __bfloat162float(bdo[j]);
            acc_db[nt] += sdb;
          }
        }
      }
    }
  }
  if (do_db) {
    for (int nt = 0; nt < n_tiles_co; ++nt) {
      float v = acc_db[nt];
      v += __shfl_xor(v, 16, 64);     // combine the 4 lq groups per co
      v += __shfl_xor(v, 32, 64);
      int co = nt * 16 + l16;
      if (lq == 0 && co < Cout)
        atomicAdd(&dbias[co], v);
    }
  }
  // accumulate into dwp (Kpad, Cout) fp32
  for (int nt = 0; nt < n_tiles_co; ++nt) {
    int co = nt * 16 + l16;
    if (co >= Cout) continue;
    for (int r = 0; r < 4; ++r) {
      int k = k_base + band + lq * 4 + r;
      if (k >= Kpad) continue;
      atomicAdd(&dwp[(long)k * Cout + co], acc[nt][r]);
    }
  }
}

// MaxPool2d(2,2) forward with packed argmax (2 bits would do; store u8).
extern "C" __global__ void maxpool2x2_fwd_kernel(
    const __hip_bfloat16* __restrict__ in,      // (N, H, W) flattened b*c
    __hip_bfloat16* __restrict__ out,           // (N, H/2, W/2)
    unsigned char* __restrict__ idx,            // (N, H/2, W/2)
    long NC, int H, int W) {
  int Ho = H / 2, Wo = W / 2;
  long total = NC * Ho * Wo;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int xo = i % Wo;
    int yo = (i / Wo) % Ho;
    long n = i / ((long)Wo * Ho);
    const __hip_bfloat16* p = in + (n * H + yo * 2) * W + xo * 2;
    float v00 = __bfloat162float(p[0]);
    float v01 = __bfloat162float(p[1]);
    float v10 = __bfloat162float(p[W]);
    float v11 = __bfloat162float(p[W + 1]);
    float m = v00;
    int a = 0;
    if (v01 > m) { m = v01; a = 1; }
    if (v10 > m) { m = v10; a = 2; }
    if (v11 > m) { m = v11; a = 3; }
    out[i] = __float2bfloat16(m);
    idx[i] = (unsigned char)a;
  }
}

// gather-style backward: each INPUT position checks its pool cell's argmax
// (no atomics, one coalesced pass — ATen's scatter-add bwd was 12 ms/step).
extern "C" __global__ void maxpool2x2_bwd_kernel(
    const __hip_bfloat16* __restrict__ dout,    // (N, H/2, W/2)
    const unsigned char* __restrict__ idx,
    __hip_bfloat16* __restrict__ din,           // (N, H, W)
    long NC, int H, int W) {
  int Ho = H / 2, Wo = W / 2;
  long total = NC * H * W;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int x = i % W;
    int y = (i / W) % H;
    long n = i / ((long)W * H);
    float g = 0.f;
    if (y / 2 < Ho && x / 2 < Wo) {
      long o = (n * Ho + y / 2) * Wo + x / 2;
      int a = (y & 1) * 2 + (x & 1);
      if (idx[o] == a) g = __bfloat162float(dout[o]);
    }
    din[i] = __float2bfloat16(g);
  }
}


// Direct conv for small Cout (< 16): one thread per output pixel, VALU
// MACs — these convs are memory-bound and MFMA n-tiles would waste >=16x.
extern "C" __global__ void conv2d_small_fwd_kernel(
    const __hip_bfloat16* __restrict__ input,   // (B, Cin, H, W)
    const __hip_bfloat16* __restrict__ wp,      // (Cout, Kpad)
    const float* __restrict__ bias,
    __hip_bfloat16* __restrict__ out,           // (B, Cout, H, W)
    int B, int Cin, int Cout, int H, int W,
    int KH, int KW, int padH, int padW, int Kpad, int relu) {
  const long HW = (long)H * W;
  const long total = (long)B * Cout * HW;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int x = i % W;
    int y = (i / W) % H;
    int co = (i / HW) % Cout;
    long b = i / (HW * Cout);
    const __hip_bfloat16* inb = input + b * Cin * HW;
    const __hip_bfloat16* wrow = wp + (long)co * Kpad;
    float acc = bias ? bias[co] : 0.f;
    int k = 0;
    for (int ci = 0; ci < Cin; ++ci) {
      const __hip_bfloat16* cib = inb + (long)ci * HW;
      for (int dy = -padH; dy < KH - padH; ++dy) {
        int yy = y + dy;
        for (int dx = -padW; dx < KW - padW; ++dx, ++k) {
          int xx = x + dx;
          if (yy >= 0 && yy < H && xx >= 0 && xx < W)
            acc += __bfloat162float(wrow[k]) *
                   __bfloat162float(cib[(long)yy * W + xx]);
        }
      }
    }
    if (relu) acc = fmaxf(acc, 0.f);
    out[i] = __float2bfloat16(acc);
  }
}


// Direct wgrad for small Cout (<= 4): dW[co,ci,dy,dx] over all pixels.
// The MFMA wgrad wastes >= 16x of its N-tile there, and its im2col
// staging re-reads the input 9x; here each input element is read once
// per block and all KH*KW*Cout partials ride in registers.
extern "C" __global__ void conv2d_wgrad_small_kernel(
    const __hip_bfloat16* __restrict__ input,   // (B, Cin, H, W)
    const __hip_bfloat16* __restrict__ dout,    // (B, Cout, H, W)
    float* __restrict__ dw,                     // (Cout, Cin, KH, KW) fp32
    int B, int Cin, int Cout, int H, int W,
    int KH, int KW, int padH, int padW, int ipb) {
  const long HW = (long)H * W;
  const int ci = blockIdx.x;
  const int b_base = blockIdx.y * ipb;
  const int kwin = KH * KW;
  float acc[4 * 9] = {};                        // [co][off]
  for (int bi = 0; bi < ipb && b_base + bi < B; ++bi) {
    const long b = b_base + bi;
    const __hip_bfloat16* cib = input + (b * Cin + ci) * HW;
    const __hip_bfloat16* dob = dout + b * Cout * HW;
    for (long p = threadIdx.x; p < HW; p += blockDim.x) {
      int y = p / W, x = p - (p / W) * W;
      float inv = __bfloat162float(cib[p]);
      if (inv == 0.f) continue;
      for (int co = 0; co < Cout; ++co) {
        const __hip_bfloat16* dco = dob + (long)co * HW;
        for (int off = 0; off < kwin; ++off) {
          // dW[co][ci][dy][dx] += in[y][x] * dout[y-dy+padH][x-dx+padW]
          int dy = off / KW - padH, dx = off - (off / KW) * KW - padW;
          int yy = y - dy, xx = x - dx;
          if (yy >= 0 && yy < H && xx >= 0 && xx < W)
            acc[co * kwin + off] += inv * __bfloat162float(dco[(long)yy * W + xx]);
        }
      }
    }
  }
  __shared__ float red[4];
  for (int co = 0; co < Cout; ++co)
    for (int off = 0; off < kwin; ++off) {
      float v = acc[co * kwin + off];
      for (int o = 1; o < 64; o <<= 1) v += __shfl_xor(v, o, 64);
      int lane = threadIdx.x & 63, wv = threadIdx.x >> 6;
      if (lane == 0) red[wv] = v;
      __syncthreads();
      if (threadIdx.x == 0) {
        float t = 0.f;
        for (int w = 0; w < (int)(blockDim.x >> 6); ++w) t += red[w];
        atomicAdd(&dw[((long)co * Cin + ci) * kwin + off], t);
      }
      __syncthreads();
    }
}


// ---------------------------------------------------------------------------
// Cout=1 3x3 s1p1 stencil pair (the LocationHead 32->1 output conv at
// (B*T, 32, 152, 160) — the only <16-channel conv in the model).
//
// v2: direct-load row accumulators.  v1 staged tiles through LDS with one
// scalar global load per element consumed behind a barrier; PMC (r2jj)
// showed both kernels 10-20x over their VALU instruction floor — pure
// latency serialization.  Here each thread owns a row item, issues its
// row reads as independent 16-40 B vector loads (the compiler keeps a
// ci-iteration's 9+ loads in flight), keeps all partials in registers,
// and never synchronizes.
extern "C" __global__ __launch_bounds__(256, 4)
void conv2d_stencil_c1_fwd_kernel(
    const __hip_bfloat16* __restrict__ input,   // (B, Cin, H, W)
    const __hip_bfloat16* __restrict__ wp,      // (1, Kpad), k = ci*9+tap
    const float* __restrict__ bias,
    __hip_bfloat16* __restrict__ out,           // (B, 1, H, W)
    int B, int Cin, int H, int W, int Kpad, int relu) {
  __shared__ __hip_bfloat16 ws[256 * 9];        // [ci][tap], Cin <= 256
  for (int e = threadIdx.x; e < Cin * 9; e += 256) ws[e] = wp[e];
  __syncthreads();
  const int CW = (W + 15) >> 4;                 // 16-px chunks per row
  const long items = (long)B * H * CW;          // item = (b, y, chunk)
  const float bb = bias ? bias[0] : 0.f;
  for (long i = (long)blockIdx.x * 256 + threadIdx.x; i < items;
       i += (long)gridDim.x * 256) {
    const int c = (int)(i % CW);
    const long q = i / CW;
    const int y = (int)(q % H);
    const long b = q / H;
    const int x0 = c * 16;
    const bool interior = (x0 >= 16) && (x0 + 18 <= W);
    float acc[16] = {};
    for (int ci = 0; ci < Cin; ++ci) {
      float wv[9];
      #pragma unroll
      for (int t = 0; t < 9; ++t) wv[t] = __bfloat162float(ws[ci * 9 + t]);
      const __hip_bfloat16* plane = input + ((b * Cin + ci) * (long)H) * W;
      #pragma unroll
      for (int r = 0; r < 3; ++r) {
        const int gy = y + r - 1;
        if (gy < 0 || gy >= H) continue;
        const __hip_bfloat16* row = plane + (long)gy * W;
        float f[18];
        if (interior) {
          __hip_bfloat16 v[20];                 // cols x0-2 .. x0+17, 40 B
          __builtin_memcpy(v, row + x0 - 2, 40);
          #pragma unroll
          for (int j = 0; j < 18; ++j) f[j] = __bfloat162float(v[j + 1]);
        } else {
          #pragma unroll
          for (int j = 0; j < 18; ++j) {
            const int xx = x0 - 1 + j;
            f[j] = (xx >= 0 && xx < W) ? __bfloat162float(row[xx]) : 0.f;
          }
        }
        #pragma unroll
        for (int dx = 0; dx < 3; ++dx)
          #pragma unroll
          for (int j = 0; j < 16; ++j)
            acc[j] += wv[r * 3 + dx] * f[j + dx];
      }
    }
    __hip_bfloat16* orow = out + (b * (long)H + y) * W + x0;
    if (x0 + 16 <= W) {
      __hip_bfloat16 hv[16];
      #pragma unroll
      for (int j = 0; j < 16; ++j) {
        float o = acc[j] + bb;
        if (relu) o = fmaxf(o, 0.f);
        hv[j] = __float2bfloat16(o);
      }
      __builtin_memcpy(orow, hv, 32);
    } else {
      for (int j = 0; j < 16 && x0 + j < W; ++j) {
        float o = acc[j] + bb;
        if (relu) o = fmaxf(o, 0.f);
        orow[j] = __float2bfloat16(o);
      }
    }
  }
}

// wgrad: item = (b, y, ci); thread ci is constant across the grid-stride
// walk (host guarantees Cin is a power of two <= 32, so 256 % Cin == 0),
// all 9 tap partials + the bias sum ride in registers for the whole
// kernel, then one shfl/LDS/global-atomic reduction per block.
extern "C" __global__ __launch_bounds__(256, 4)
void conv2d_stencil_c1_wgrad_kernel(
    const __hip_bfloat16* __restrict__ input,   // (B, Cin, H, W)
    const __hip_bfloat16* __restrict__ dout,    // (B, 1, H, W)
    float* __restrict__ dw,                     // (1, Cin, 3, 3) fp32
    float* __restrict__ dbias,                  // (1) fp32 or nullptr
    int B, int Cin, int H, int W) {
  __shared__ float reds[32 * 9];
  __shared__ float redb;
  const int tid = threadIdx.x;
  const int ci = tid & (Cin - 1);
  float acc[9] = {};
  float accb = 0.f;
  const long items = (long)B * H * Cin;         // item = (b, y, ci)
  for (long i = (long)blockIdx.x * 256 + tid; i < items;
       i += (long)gridDim.x * 256) {
    const long q = i / Cin;                     // i % Cin == ci
    const int y = (int)(q % H);
    const long b = q / H;
    const __hip_bfloat16* xrow = input + ((b * Cin + ci) * (long)H + y) * W;
    const __hip_bfloat16* dbase = dout + b * (long)H * W;
    for (int x0 = 0; x0 < W; x0 += 16) {
      float xf[16];
      if (x0 + 16 <= W) {
        __hip_bfloat16 v[16];
        __builtin_memcpy(v, xrow + x0, 32);
        #pragma unroll
        for (int j = 0; j < 16; ++j) xf[j] = __bfloat162float(v[j]);
      } else {
        #pragma unroll
        for (int j = 0; j < 16; ++j)
          xf[j] = (x0 + j < W) ? __bfloat162float(xrow[x0 + j]) : 0.f;
      }
      const bool interior = (x0 >= 16) && (x0 + 18 <= W);
      // dW[ty,tx] += x[y][x] * dy[y+1-ty][x+1-tx]
      #pragma unroll
      for (int ty = 0; ty < 3; ++ty) {
        const int gy = y + 1 - ty;
        if (gy < 0 || gy >= H) continue;
        const __hip_bfloat16* drow = dbase + (long)gy * W;
        float f[18];                            // dy cols x0-1 .. x0+16
        if (interior) {
          __hip_bfloat16 v[20];
          __builtin_memcpy(v, drow + x0 - 2, 40);
          #pragma unroll
          for (int j = 0; j < 18; ++j) f[j] = __bfloat162float(v[j + 1]);
        } else {
          #pragma unroll
          for (int j = 0; j < 18; ++j) {
            const int xx = x0 - 1 + j;
            f[j] = (xx >= 0 && xx < W) ? __bfloat162float(drow[xx]) : 0.f;
          }
        }
        #pragma unroll
        for (int tx = 0; tx < 3; ++tx)
          #pragma unroll
          for (int j = 0; j < 16; ++j)
            acc[ty * 3 + tx] += xf[j] * f[j + 2 - tx];
        if (ty == 1 && ci == 0) {
          #pragma unroll
          for (int j = 0; j < 16; ++j) accb += f[j + 1];
        }
      }
    }
  }
  // lanes l and l^Cin .. share ci; fold strips above Cin
  #pragma unroll
  for (int o = 32; o >= 1; o >>= 1) {
    if (o >= Cin) {
      #pragma unroll
      for (int k = 0; k < 9; ++k) acc[k] += __shfl_xor(acc[k], o, 64);
      accb += __shfl_xor(accb, o, 64);
    }
  }
  for (int e = tid; e < Cin * 9; e += 256) reds[e] = 0.f;   // Cin*9 can be
  if (tid == 0) redb = 0.f;                                 // > blockDim
  __syncthreads();
  if ((tid & 63) < Cin) {                       // lane ci of each wave
    #pragma unroll
    for (int k = 0; k < 9; ++k) atomicAdd(&reds[ci * 9 + k], acc[k]);
    if (ci == 0) atomicAdd(&redb, accb);
  }
  __syncthreads();
  for (int e = tid; e < Cin * 9; e += 256) atomicAdd(&dw[e], reds[e]);
  if (tid == 0 && dbias) atomicAdd(dbias, redb);
}

// ---------------------------------------------------------------------------
// Small-HW conv path (the 19x20 ResBlock/GatedResBlock stack, HW = 380):
// at 20-wide rows the in-kernel im2col gather of the MFMA kernels
// degenerates (runs <= 20, halo logic per element), measured ~8x off the
// HBM floor.  For these shapes conv is materialized im2col -> batched
// hipBLASLt GEMM (a plain library GEMM per the design rules) with these
// two hand-written movement kernels; col2im is the gather formulation
// (each dx element sums its 9 tap sources - no atomics).

// col[b][ci*9 + (dy*3+dx)][p] = x[b][ci][y+dy-1][x+dx-1]  (3x3, pad 1)
extern "C" __global__ __launch_bounds__(256)
void im2col_3x3_kernel(
    const __hip_bfloat16* __restrict__ x,   // (B, C, H, W)
    __hip_bfloat16* __restrict__ col,       // (B, C*9, H*W)
    long BC, int H, int W) {                // BC = B*C planes
  const int HW = H * W;
  // one block per (plane, tap): writes one contiguous (H*W) run of col
  const long plane = blockIdx.x;
  const int tap = blockIdx.y;               // 0..8
  if (plane >= BC) return;
  const int dy = tap / 3 - 1, dx = tap % 3 - 1;
  const __hip_bfloat16* xp = x + plane * (long)HW;
  __hip_bfloat16* cp = col + (plane * 9 + tap) * (long)HW;
  const __hip_bfloat16 z = __float2bfloat16(0.f);
  for (int p = threadIdx.x; p < HW; p += 256) {
    int y = p / W, xx = p - y * W;
    int sy = y + dy, sx = xx + dx;
    cp[p] = (sy >= 0 && sy < H && sx >= 0 && sx < W) ? xp[sy * W + sx] : z;
  }
}

// dx[b][ci][y][x] = sum_tap dcol[b][ci*9+tap][(y-dy)(x-dx)]
extern "C" __global__ __launch_bounds__(256)
void col2im_3x3_kernel(
    const __hip_bfloat16* __restrict__ dcol,  // (B, C*9, H*W)
    __hip_bfloat16* __restrict__ dx,          // (B, C, H, W)
    long BC, int H, int W) {
  const int HW = H * W;
  const long plane = blockIdx.x;
  if (plane >= BC) return;
  const __hip_bfloat16* cp = dcol + plane * 9 * (long)HW;
  __hip_bfloat16* xp = dx + plane * (long)HW;
  for (int p = threadIdx.x; p < HW; p += 256) {
    int y = p / W, xx = p - y * W;
    float acc = 0.f;
    #pragma unroll
    for (int tap = 0; tap < 9; ++tap) {
      int dy = tap / 3 - 1, dx_ = tap % 3 - 1;
      int sy = y - dy, sx = xx - dx_;
      if (sy >= 0 && sy < H && sx >= 0 && sx < W)
        acc += __bfloat162float(cp[tap * (long)HW + sy * W + sx]);
    }
    xp[p] = __float2bfloat16(acc);
  }
}


// Vectorized maxpool pair for W % 8 == 0 (all pools except the 38x40 one):
// bwd was one scalar 2B store per input element with a div/mod each
// (measured 7.5 ms/step); here a thread expands 4 pool cells into one 16B
// row write.  fwd reads two 16-px input rows as 32B vectors.
extern "C" __global__ void maxpool2x2_fwd_vec_kernel(
    const __hip_bfloat16* __restrict__ in,      // (NC, H, W)
    __hip_bfloat16* __restrict__ out,           // (NC, H/2, W/2)
    unsigned char* __restrict__ idx,
    long NC, int H, int W) {
  const int Ho = H / 2, Wo = W / 2;
  const int WG = Wo / 8;                        // 8 output px per thread
  const long total = NC * Ho * WG;
  for (long g = (long)blockIdx.x * blockDim.x + threadIdx.x; g < total;
       g += (long)gridDim.x * blockDim.x) {
    int xo = (int)(g % WG) * 8;
    int yo = (int)((g / WG) % Ho);
    long n = g / ((long)WG * Ho);
    const __hip_bfloat16* p = in + (n * H + yo * 2) * (long)W + xo * 2;
    __hip_bfloat16 r0[16], r1[16], ov[8];
    unsigned char iv[8];
    __builtin_memcpy(r0, p, 32);
    __builtin_memcpy(r1, p + W, 32);
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float v00 = __bfloat162float(r0[2 * j]);
      float v01 = __bfloat162float(r0[2 * j + 1]);
      float v10 = __bfloat162float(r1[2 * j]);
      float v11 = __bfloat162float(r1[2 * j + 1]);
      float m = v00; int a = 0;
      if (v01 > m) { m = v01; a = 1; }
      if (v10 > m) { m = v10; a = 2; }
      if (v11 > m) { m = v11; a = 3; }
      ov[j] = __float2bfloat16(m);
      iv[j] = (unsigned char)a;
    }
    long o = (n * Ho + yo) * (long)Wo + xo;
    __builtin_memcpy(out + o, ov, 16);
    __builtin_memcpy(idx + o, iv, 8);
  }
}

extern "C" __global__ void maxpool2x2_bwd_vec_kernel(
    const __hip_bfloat16* __restrict__ dout,    // (NC, H/2, W/2)
    const unsigned char* __restrict__ idx,
    __hip_bfloat16* __restrict__ din,           // (NC, H, W)
    long NC, int H, int W) {
  const int Ho = H / 2, Wo = W / 2;
  const int WG = W / 8;                         // 8 input px = 4 cells
  const long total = NC * H * WG;
  const __hip_bfloat16 z = __float2bfloat16(0.f);
  for (long g = (long)blockIdx.x * blockDim.x + threadIdx.x; g < total;
       g += (long)gridDim.x * blockDim.x) {
    int xg = (int)(g % WG) * 8;
    int y = (int)((g / WG) % H);
    long n = g / ((long)WG * H);
    long o = (n * Ho + (y >> 1)) * (long)Wo + (xg >> 1);
    __hip_bfloat16 dv[4], ov[8];
    unsigned char iv[4];
    __builtin_memcpy(dv, dout + o, 8);
    __builtin_memcpy(iv, idx + o, 4);
    const int row2 = (y & 1) * 2;
    #pragma unroll
    for (int c = 0; c < 4; ++c) {
      ov[2 * c] = (iv[c] == row2) ? dv[c] : z;
      ov[2 * c + 1] = (iv[c] == row2 + 1) ? dv[c] : z;
    }
    __builtin_memcpy(din + (n * H + y) * (long)W + xg, ov, 16);
  }
}


// ---------------------------------------------------------------------------
// Windowed small-image variants (3x3 s1p1, (H+2)*(W+2) <= 484 — the 19x20
// ResBlock/GatedResBlock stack).  The generic kernels gather im2col
// elements straight from global with per-element decode + bounds; at
// W = 20 that staging is instruction- and latency-bound (PMC r2jj: the
// wgrad ran at 12k VALU + 11k SALU per wave).  Here each block first
// copies the PADDED input window of the <= 15 ci it needs into LDS
// (coalesced, zero-filled halo), then builds im2col tiles from LDS with
// NO bounds checks (padding makes every read valid) and one decode per
// (ci,dy) TRIPLE (dx=0,1,2 are shifted reads of the same row).

#define SWHP 22                    // max padded H
#define SWWP 22                    // max padded W

// stage the padded window of ci [ci0, ci0+nci) of image b
__device__ __forceinline__ void stage_window(
    __hip_bfloat16* win, const __hip_bfloat16* __restrict__ input,
    long b, int Cin, int ci0, int nci, int H, int W, int tid) {
  const int Hp = H + 2, Wp = W + 2;
  const int nwin = nci * Hp * Wp;
  const __hip_bfloat16 z = __float2bfloat16(0.f);
  // 4-batched: the load of iteration i must not serialize behind the
  // LDS store of iteration i-1 (the stencil-v1 lesson, PMC r2jj)
  for (int e0 = tid; e0 < nwin; e0 += 1024) {
    __hip_bfloat16 v[4];
    #pragma unroll
    for (int u = 0; u < 4; ++u) {
      const int e = e0 + u * 256;
      v[u] = z;
      if (e < nwin) {
        const int c = e % Wp;
        const int t = e / Wp;
        const int r = t % Hp;
        const int ic = t / Hp;
        const int gy = r - 1, gx = c - 1;
        if (gy >= 0 && gy < H && gx >= 0 && gx < W)
          v[u] = input[((b * Cin + ci0 + ic) * (long)H + gy) * W + gx];
      }
    }
    #pragma unroll
    for (int u = 0; u < 4; ++u)
      if (e0 + u * 256 < nwin) win[e0 + u * 256] = v[u];
  }
}

// build one [k][px] (transpose=0: [px][k]) im2col chunk tile from the
// window: k rows [k_base, k_base+krows), px [p0, p0+npx)
__device__ __forceinline__ void build_tile_from_window(
    char* lds, const __hip_bfloat16* win, int ci0,
    int k_base, int krows, int p0, int npx, int K_real,
    int H, int W, int wrecip, int transpose, int tid) {
  const int Hp = H + 2, Wp = W + 2;
  const int HW = H * W;
  const int t0 = k_base / 3;                    // triple = ci*3 + dy
  const int t1 = (min(k_base + krows, K_real) + 2) / 3;
  const int ntrip = t1 - t0;
  const int ngrp = npx / 8;
  const __hip_bfloat16 z = __float2bfloat16(0.f);
  for (int task = tid; task < ntrip * ngrp; task += 256) {
    const int g = task / ntrip;
    const int tr = task - g * ntrip + t0;
    const int ci = (tr * 21846) >> 16;          // tr / 3 (tr < 3*128*3)
    const int dyy = tr - ci * 3;
    const __hip_bfloat16* w0 = win + (ci - ci0) * Hp * Wp;
    const int pbase = p0 + g * 8;
    const int y0p = (int)(((long)pbase * wrecip) >> 20);
    const int x0p = pbase - y0p * W;
    __hip_bfloat16 v3[3][8];
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int wrap = (x0p + j >= W) ? 1 : 0;
      const int yj = y0p + wrap;
      const int xj = x0p + j - wrap * W;
      const bool ok = (pbase + j < HW);
      const int off = ok ? (yj + dyy) * Wp + xj : 0;
      #pragma unroll
      for (int dx = 0; dx < 3; ++dx)
        v3[dx][j] = ok ? w0[off + dx] : z;
    }
    #pragma unroll
    for (int dx = 0; dx < 3; ++dx) {
      const int k = tr * 3 + dx;
      const int kk = k - k_base;
      if (kk < 0 || kk >= krows || k >= K_real) continue;
      if (transpose) {                          // [k][px] (wgrad A^T)
        __builtin_memcpy(lds + cswz(kk, g * 8), v3[dx], 16);
      } else {                                  // [px][k] (fwd A)
        #pragma unroll
        for (int j = 0; j < 8; ++j)
          *(__hip_bfloat16*)(lds + cswz(g * 8 + j, kk)) = v3[dx][j];
      }
    }
  }
}

extern "C" __global__ __launch_bounds__(256, 4)
void conv2d_fwd_smallhw_kernel(
    const __hip_bfloat16* __restrict__ input,   // (B, Cin, H, W)
    const __hip_bfloat16* __restrict__ wp,      // (Cout, Kpad)
    const float* __restrict__ bias,
    __hip_bfloat16* __restrict__ out,           // (B, Cout, H, W)
    int B, int Cin, int Cout, int H, int W, int Kpad, int relu) {
  const int HW = H * W;
  const int p0 = blockIdx.x * CTILE;
  const int n0 = blockIdx.y * 128;
  const long b = blockIdx.z;
  if (p0 >= HW || n0 >= Cout) return;
  const int K_real = Cin * 9;
  const int wrecip = (1048576 + W - 1) / W;
  extern __shared__ char lds[];                 // [64px][128k] 16 KB
  __shared__ __hip_bfloat16 win[15 * SWHP * SWWP];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int l16 = lane & 15;
  const int lq = lane >> 4;
  const int band = wave * 16;
  const int tid = threadIdx.x;
  f32x4c acc[2][4];
  for (int h = 0; h < 2; ++h)
    for (int nt = 0; nt < 4; ++nt) acc[h][nt] = (f32x4c){0, 0, 0, 0};
  const int k_hi = (K_real + 31) / 32 * 32;
  for (int k0 = 0; k0 < k_hi; k0 += KC) {
    const int krows = min(k_hi - k0, KC);
    const int ci0 = k0 / 9;
    const int ci1 = min((min(k0 + krows, K_real) - 1) / 9, Cin - 1);
    __syncthreads();
    stage_window(win, input, b, Cin, ci0, ci1 - ci0 + 1, H, W, tid);
    __syncthreads();
    build_tile_from_window(lds, win, ci0, k0, krows, p0, CTILE,
                           K_real, H, W, wrecip, 0, tid);
    __syncthreads();
    const int ks_count = krows / 32;
    for (int ks = 0; ks < ks_count; ++ks) {
      bf16x8c bi[4];
      for (int nt = 0; nt < 4; ++nt)
        bi[nt] = clds8(lds, cswz(nt * 16 + l16, ks * 32 + lq * 8));
      for (int h = 0; h < 2; ++h) {
        if (n0 + h * 64 + band >= Cout) continue;
        int co_a = n0 + h * 64 + band + l16;
        bf16x8c a = (co_a < Cout)
            ? *(const bf16x8c*)(wp + (long)co_a * Kpad + k0 + ks * 32 + lq * 8)
            : (bf16x8c){0, 0, 0, 0, 0, 0, 0, 0};
        for (int nt = 0; nt < 4; ++nt)
          acc[h][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a, bi[nt], acc[h][nt], 0, 0, 0);
      }
    }
  }
  for (int h = 0; h < 2; ++h) {
    for (int r = 0; r < 4; ++r) {
      int co = n0 + h * 64 + band + lq * 4 + r;
      if (co >= Cout) continue;
      float bv = bias ? bias[co] : 0.f;
      for (int nt = 0; nt < 4; ++nt) {
        int p = p0 + nt * 16 + l16;
        if (p >= HW) continue;
        float v = acc[h][nt][r] + bv;
        if (relu) v = fmaxf(v, 0.f);
        out[(b * Cout + co) * HW + p] = __float2bfloat16(v);
      }
    }
  }
}

extern "C" __global__ __launch_bounds__(256, 2)
void conv2d_wgrad_smallhw_kernel(
    const __hip_bfloat16* __restrict__ input,   // (B, Cin, H, W)
    const __hip_bfloat16* __restrict__ dout,    // (B, Cout, H, W)
    float* __restrict__ dwp,                    // (Kpad, Cout) fp32
    float* __restrict__ dbias,                  // (Cout) fp32 or nullptr
    int B, int Cin, int Cout, int H, int W, int Kpad, int ipb) {
  const int HW = H * W;
  const int k_base = blockIdx.x * CTILE;        // 64 k-rows of dW
  const int b_base = blockIdx.z * ipb;
  const int K_real = Cin * 9;
  if (k_base >= K_real) return;
  const int n_tiles_co = (Cout + 15) / 16;
  const int wrecip = (1048576 + W - 1) / W;
  extern __shared__ char lds[];                 // [64k][128px] 16 KB
  __shared__ __hip_bfloat16 win[9 * SWHP * SWWP];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int l16 = lane & 15;
  const int lq = lane >> 4;
  const int band = wave * 16;
  const int tid = threadIdx.x;
  f32x4c acc[8];
  for (int nt = 0; nt < 8; ++nt) acc[nt] = (f32x4c){0, 0, 0, 0};
  const bool do_db = dbias && blockIdx.x == 0 && wave == 0;
  float acc_db[8] = {};
  const int ci0 = k_base / 9;
  const int ci1 = min((min(k_base + CTILE, K_real) - 1) / 9, Cin - 1);
  for (int bi = 0; bi < ipb && b_base + bi < B; ++bi) {
    const long b = b_base + bi;
    const __hip_bfloat16* dob = dout + b * Cout * (long)HW;
    __syncthreads();
    stage_window(win, input, b, Cin, ci0, ci1 - ci0 + 1, H, W, tid);
    for (int p0 = 0; p0 < HW; p0 += KC) {
      __syncthreads();
      build_tile_from_window(lds, win, ci0, k_base, CTILE, p0, KC,
                             K_real, H, W, wrecip, 1, tid);
      __syncthreads();
      for (int ks = 0; ks < 4; ++ks) {
        bf16x8c a = clds8(lds, cswz(band + l16, ks * 32 + lq * 8));
        for (int nt = 0; nt < n_tiles_co; ++nt) {
          int co = nt * 16 + l16;
          bf16x8c bdo = (bf16x8c){0, 0, 0, 0, 0, 0, 0, 0};
          if (co < Cout) {
            int p = p0 + ks * 32 + lq * 8;
            if (p + 8 <= HW) {
              __builtin_memcpy(&bdo, dob + (long)co * HW + p, 16);
            } else {
              for (int j = 0; j < 8; ++j)
                if (p + j < HW)
                  bdo[j] = ((const __bf16*)dob)[(long)co * HW + p + j];
            }
          }
          acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bdo, acc[nt],
                                                            0, 0, 0);
          if (do_db) {
            float sdb = 0.f;
            #pragma unroll
            for (int j = 0; j < 8; ++j) sdb += __bfloat162float(bdo[j]);
            acc_db[nt] += sdb;
          }
        }
      }
    }
  }
  for (int nt = 0; nt < n_tiles_co; ++nt) {
    for (int r = 0; r < 4; ++r) {
      int k = k_base + band + lq * 4 + r;
      int co = nt * 16 + l16;
      if (k < Kpad && co < Cout && k < K_real)
        atomicAdd(&dwp[(long)k * Cout + co], acc[nt][r]);
    }
    if (do_db) {
      int co = nt * 16 + l16;
      if (co < Cout) atomicAdd(&dbias[co], acc_db[nt]);
    }
  }
}
