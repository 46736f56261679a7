// Bilinear 2x upsample (align_corners=false), forward + backward — the
// LocationHead's 3-stage upsample chain.
//
// PyTorch's generic NCHW bilinear kernel ran fp32/scalar and was 29% of the
// whole SL step (profiles/r01_notes.md).  For the fixed scale-2,
// align_corners=false case every output pixel is a 4-tap filter with weights
// from {9/16, 3/16, 1/16} and source indices derivable with shifts:
//   src = (dst + 0.5)/2 - 0.5  ->  x0 = (dst-1)>>1, frac in {0.25, 0.75}.
//
// Memory-bound: one thread per output pixel over (N*C, H2, W2) with
// row-contiguous coalesced access; fp32 math, fp32 or bf16 storage.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

using bf16 = __hip_bfloat16;

template <typename T>
__device__ inline float ld(const T* p);
template <> __device__ inline float ld<float>(const float* p) { return *p; }
template <> __device__ inline float ld<bf16>(const bf16* p) { return __bfloat162float(*p); }

template <typename T>
__device__ inline void st(T* p, float v);
template <> __device__ inline void st<float>(float* p, float v) { *p = v; }
template <> __device__ inline void st<bf16>(bf16* p, float v) { *p = __float2bfloat16(v); }

// dst index -> (src0, src1, w0, w1) along one axis for scale-2,
// align_corners=false, with edge clamping.
__device__ inline void taps2x(int d, int n_src, int& s0, int& s1, float& w0) {
  // src = (d + 0.5f) * 0.5f - 0.5f
  float src = fmaf((float)d, 0.5f, -0.25f);
  float floor_src = floorf(src);
  s0 = (int)floor_src;
  float frac = src - floor_src;          // 0.25 or 0.75
  s1 = s0 + 1;
  if (s0 < 0) s0 = 0;
  if (s1 > n_src - 1) s1 = n_src - 1;
  w0 = 1.f - frac;
}

// One thread per 8 consecutive output pixels: the 64-bit div/mod and the
// y-taps amortize 8x (at 1.6G output elements the per-element divisions
// dominated), and the 8 stores become one 16B write for bf16.
template <typename T>
__device__ void upsample2x_fwd(const T* __restrict__ in, T* __restrict__ out,
                               int NC, int H, int W) {
  const int H2 = H * 2, W2 = W * 2;     // W2 is a multiple of 8 here
  const int WG = W2 / 8;
  const long total_g = (long)NC * H2 * WG;
  for (long g = (long)blockIdx.x * blockDim.x + threadIdx.x; g < total_g;
       g += (long)gridDim.x * blockDim.x) {
    int xg = (int)(g % WG) * 8;
    int y = (int)((g / WG) % H2);
    long nc = g / ((long)WG * H2);
    int y0, y1;
    float wy0;
    taps2x(y, H, y0, y1, wy0);
    const T* r0 = in + nc * (long)H * W + (long)y0 * W;
    const T* r1 = in + nc * (long)H * W + (long)y1 * W;
    T vals[8];
    for (int j = 0; j < 8; ++j) {
      int x0, x1;
      float wx0;
      taps2x(xg + j, W, x0, x1, wx0);
      float v = wy0 * (wx0 * ld(r0 + x0) + (1.f - wx0) * ld(r0 + x1)) +
                (1.f - wy0) * (wx0 * ld(r1 + x0) + (1.f - wx0) * ld(r1 + x1));
      st(vals + j, v);
    }
    __builtin_memcpy(out + (nc * (long)H2 + y) * W2 + xg, vals, sizeof(vals));
  }
}

// backward: each SOURCE pixel gathers from the (at most) 16 destination
// pixels whose stencil touches it — gather, not atomics.  For scale 2 each
// source pixel is touched by a fixed 4x4 window of outputs.
template <typename T>
__device__ void upsample2x_bwd(const T* __restrict__ gout, T* __restrict__ gin,
                               int NC, int H, int W) {
  const int H2 = H * 2, W2 = W * 2;     // W is a multiple of 4 here
  const int WG = W / 4;
  const long total_g = (long)NC * H * WG;
  for (long g = (long)blockIdx.x * blockDim.x + threadIdx.x; g < total_g;
       g += (long)gridDim.x * blockDim.x) {
    int sxg = (int)(g % WG) * 4;
    int sy = (int)((g / WG) % H);
    long nc = g / ((long)WG * H);
    const T* base = gout + nc * (long)H2 * W2;
    T outv[4];
    for (int j = 0; j < 4; ++j) {
    int sx = sxg + j;
    float acc = 0.f;
    // destination rows/cols that can reference (sy, sx): d in [2s-1, 2s+2]
    for (int dy = sy * 2 - 1; dy <= sy * 2 + 2; ++dy) {
      if (dy < 0 || dy >= H2) continue;
      int y0, y1; float wy0;
      taps2x(dy, H, y0, y1, wy0);
      float wy = (y0 == sy ? wy0 : 0.f) + (y1 == sy ? 1.f - wy0 : 0.f);
      if (wy == 0.f) continue;
      for (int dx = sx * 2 - 1; dx <= sx * 2 + 2; ++dx) {
        if (dx < 0 || dx >= W2) continue;
        int x0, x1; float wx0;
        taps2x(dx, W, x0, x1, wx0);
        float wx = (x0 == sx ? wx0 : 0.f) + (x1 == sx ? 1.f - wx0 : 0.f);
        if (wx == 0.f) continue;
        acc += wy * wx * ld(base + (long)dy * W2 + dx);
      }
    }
    st(outv + j, acc);
    }
    __builtin_memcpy(gin + (nc * (long)H + sy) * W + sxg, outv, sizeof(outv));
  }
}

// Fast fwd: scale-2 align_corners=false weights are constant by parity
// (out[2s] = .25 in[s-1] + .75 in[s]; out[2s+1] = .75 in[s] + .25 in[s+1],
// separable in y) — the generic kernel burned a taps2x() per output px.
// Interior items first, border items after (uniform waves, as in bwd).
extern "C" __global__ void upsample2x_fwd_bf16_fast(
    const bf16* __restrict__ in, bf16* __restrict__ out,
    int NC, int H, int W) {
  const int H2 = H * 2, W2 = W * 2;
  const int WG = W2 / 8;                        // 8 out px = 4 src pairs
  // interior: out rows 2..H2-3 (src rows fully valid), out col groups
  // whose src cols s-1..s+4 stay in range: groups 1..WG-2
  const int IW = WG - 2, IH = H2 - 4;
  const long total_i = (IW > 0 && IH > 0) ? (long)NC * IH * IW : 0;
  const int bc = 4 * WG + 2 * (H2 - 4);         // border groups per plane
  const long total = total_i + (long)NC * bc;
  for (long g = (long)blockIdx.x * blockDim.x + threadIdx.x; g < total;
       g += (long)gridDim.x * blockDim.x) {
    int y, xg;
    long nc;
    bool interior;
    if (g < total_i) {
      xg = ((int)(g % IW) + 1) * 8;
      y = (int)((g / IW) % IH) + 2;
      nc = g / ((long)IW * IH);
      interior = true;
    } else {
      const long gb = g - total_i;
      nc = gb / bc;
      const int r = (int)(gb - nc * bc);
      if (r < 2 * WG) { y = r / WG; xg = (r % WG) * 8; }
      else if (r < 4 * WG) { y = H2 - 2 + (r - 2 * WG) / WG;
                             xg = ((r - 2 * WG) % WG) * 8; }
      else { const int r2 = r - 4 * WG; y = 2 + (r2 >> 1);
             xg = (r2 & 1) ? (WG - 1) * 8 : 0; }
      interior = false;
    }
    const bf16* base = in + nc * (long)H * W;
    if (interior) {
      // src rows y0,y1 for out row y; weights by parity
      const int s0 = (y - 1) >> 1;
      const float wy0 = (y & 1) ? 0.75f : 0.25f;   // weight of row s0
      const int sx = (xg - 1) >> 1;                // first src col needed
      bf16 r0[6], r1[6];
      __builtin_memcpy(r0, base + (long)s0 * W + sx, 12);
      __builtin_memcpy(r1, base + (long)(s0 + 1) * W + sx, 12);
      float f[6];
      #pragma unroll
      for (int t = 0; t < 6; ++t)
        f[t] = wy0 * __bfloat162float(r0[t]) +
               (1.f - wy0) * __bfloat162float(r1[t]);
      bf16 ov[8];
      #pragma unroll
      for (int j = 0; j < 4; ++j) {
        // out cols xg+2j (even), xg+2j+1 (odd); src col of even = s
        // f index: src col (xg+2j-1)>>1 ... relative to sx
        const int fe = ((xg + 2 * j - 1) >> 1) - sx;  // = j if xg odd-base
        ov[2 * j] = __float2bfloat16(0.25f * f[fe] + 0.75f * f[fe + 1]);
        ov[2 * j + 1] = __float2bfloat16(0.75f * f[fe + 1] +
                                         0.25f * f[fe + 2]);
      }
      __builtin_memcpy(out + (nc * (long)H2 + y) * W2 + xg, ov, 16);
    } else {
      bf16 vals[8];
      int y0, y1;
      float wy0;
      taps2x(y, H, y0, y1, wy0);
      const bf16* r0 = base + (long)y0 * W;
      const bf16* r1 = base + (long)y1 * W;
      for (int j = 0; j < 8; ++j) {
        int x0, x1;
        float wx0;
        taps2x(xg + j, W, x0, x1, wx0);
        float v = wy0 * (wx0 * ld(r0 + x0) + (1.f - wx0) * ld(r0 + x1)) +
                  (1.f - wy0) * (wx0 * ld(r1 + x0) + (1.f - wx0) * ld(r1 + x1));
        vals[j] = __float2bfloat16(v);
      }
      __builtin_memcpy(out + (nc * (long)H2 + y) * W2 + xg, vals, 16);
    }
  }
}

extern "C" __global__ void upsample2x_fwd_f32(const float* in, float* out,
                                              int NC, int H, int W) {
  upsample2x_fwd<float>(in, out, NC, H, W);
}
extern "C" __global__ void upsample2x_fwd_bf16(const bf16* in, bf16* out,
                                               int NC, int H, int W) {
  upsample2x_fwd<bf16>(in, out, NC, H, W);
}
extern "C" __global__ void upsample2x_bwd_f32(const float* gout, float* gin,
                                              int NC, int H, int W) {
  upsample2x_bwd<float>(gout, gin, NC, H, W);
}
extern "C" __global__ void upsample2x_bwd_bf16(const bf16* gout, bf16* gin,
                                               int NC, int H, int W) {
  upsample2x_bwd<bf16>(gout, gin, NC, H, W);
}

// Fast interior path: for scale 2 / align_corners=false the tap fractions
// are ALWAYS 0.25 or 0.75, so each interior source pixel gathers a fixed
// 4x4 dest window with constant separable weights [.25,.75,.75,.25] —
// no floorf, no per-tap branches (the generic kernel above spent ~16
// taps2x() per source px and ran ~10x off the HBM floor).  Border sources
// (first/last row/col: the clamped taps add 0.25 extra weight) and ragged
// groups fall back to the generic per-pixel path.
template <typename T>
__device__ void upsample2x_bwd_onepx(const T* __restrict__ base, T* gin,
                                     long nc, int sy, int sx, int H, int W) {
  const int H2 = H * 2, W2 = W * 2;
  float acc = 0.f;
  for (int dy = sy * 2 - 1; dy <= sy * 2 + 2; ++dy) {
    if (dy < 0 || dy >= H2) continue;
    int y0, y1; float wy0;
    taps2x(dy, H, y0, y1, wy0);
    float wy = (y0 == sy ? wy0 : 0.f) + (y1 == sy ? 1.f - wy0 : 0.f);
    if (wy == 0.f) continue;
    for (int dx = sx * 2 - 1; dx <= sx * 2 + 2; ++dx) {
      if (dx < 0 || dx >= W2) continue;
      int x0, x1; float wx0;
      taps2x(dx, W, x0, x1, wx0);
      float wx = (x0 == sx ? wx0 : 0.f) + (x1 == sx ? 1.f - wx0 : 0.f);
      if (wx == 0.f) continue;
      acc += wy * wx * ld(base + (long)dy * W2 + dx);
    }
  }
  st(gin + (nc * (long)H + sy) * W + sx, acc);
}

extern "C" __global__ void upsample2x_bwd_bf16_fast(
    const bf16* __restrict__ gout, bf16* __restrict__ gin,
    int NC, int H, int W) {
  const int H2 = H * 2, W2 = W * 2;
  const int WG = W / 4;
  const long total_g = (long)NC * H * WG;
  for (long g = (long)blockIdx.x * blockDim.x + threadIdx.x; g < total_g;
       g += (long)gridDim.x * blockDim.x) {
    const int sxg = (int)(g % WG) * 4;
    const int sy = (int)((g / WG) % H);
    const long nc = g / ((long)WG * H);
    const bf16* base = gout + nc * (long)H2 * W2;
    if (sy == 0 || sy == H - 1 || sxg == 0 || sxg + 4 >= W) {
      for (int j = 0; j < 4; ++j)
        upsample2x_bwd_onepx(base, gin, nc, sy, sxg + j, H, W);
      continue;
    }
    // interior: dest rows 2sy-1..2sy+2, dest cols 2sxg-1..2sxg+8 — read 12
    // (24 B, 4B-aligned since sxg is even) and use elements 1..10
    const float wy[4] = {0.25f, 0.75f, 0.75f, 0.25f};
    float acc[4] = {};
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      bf16 v[12];
      __builtin_memcpy(v, base + (long)(sy * 2 - 1 + i) * W2 + sxg * 2 - 2, 24);
      float f[12];
      #pragma unroll
      for (int t = 1; t < 12; ++t) f[t] = __bfloat162float(v[t]);
      #pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[j] += wy[i] * (0.25f * (f[2 * j + 1] + f[2 * j + 4]) +
                           0.75f * (f[2 * j + 2] + f[2 * j + 3]));
    }
    bf16 outv[4];
    #pragma unroll
    for (int j = 0; j < 4; ++j) outv[j] = __float2bfloat16(acc[j]);
    __builtin_memcpy(gin + (nc * (long)H + sy) * W + sxg, outv, 8);
  }
}
