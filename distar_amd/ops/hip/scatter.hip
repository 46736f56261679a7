// K3: entity->map scatter connection (reference module_utils.py:11-34).
//
// Writes entity embeddings straight into the NCHW map: out[b][c][y][x] +=
// src[b][n][c] for every valid entity.  The eager formulation (index_add on
// a (B*HW, C) row buffer + permute().contiguous()) moves ~10 GB per SL step
// at bench shapes; here the only full-map traffic is the unavoidable
// zero-init, and the scatter itself is one packed-bf16 atomic per
// (entity, channel) — gfx950's global_atomic_pk_add_bf16 via HIP's
// unsafeAtomicAdd on __hip_bfloat162, with the neighbor lane of the pair
// receiving +0 (x is paired along W, so the pair address is 4B-aligned).
//
// 'add' semantics only — 'cover' (unused by the default configs) stays on
// the eager path.  Backward is a plain gather: dsrc[b][n][c] =
// dout[b][c][y][x].
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

extern "C" __global__ void scatter_add_bf16_kernel(
    const __hip_bfloat16* __restrict__ src,   // (B, N, C)
    const int* __restrict__ xy,               // (B, N, 2) clamped ints
    const int* __restrict__ entity_num,       // (B,) or nullptr
    __hip_bfloat16* __restrict__ out,         // (B, C, H, W) zero-inited
    int B, int N, int C, int H, int W) {
  const long total = (long)B * N * C;
  const long HW = (long)H * W;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int c = i % C;
    long bn = i / C;
    int n = bn % N;
    long b = bn / N;
    if (entity_num && n >= entity_num[b]) continue;
    int x = xy[bn * 2 + 0];
    int y = xy[bn * 2 + 1];
    __hip_bfloat16 v = src[i];
    if (__bfloat162float(v) == 0.f) continue;        // masked/zero rows
    long base = (b * C + c) * HW + (long)y * W;
    // pair along W: address must be 4B aligned for the packed atomic
    long pair = base + (x & ~1);
    __hip_bfloat162 add;
    if (x & 1) { add.x = __float2bfloat16(0.f); add.y = v; }
    else       { add.x = v; add.y = __float2bfloat16(0.f); }
    unsafeAtomicAdd(reinterpret_cast<__hip_bfloat162*>(out + pair), add);
  }
}

extern "C" __global__ void scatter_add_bf16_bwd_kernel(
    const __hip_bfloat16* __restrict__ dout,  // (B, C, H, W)
    const int* __restrict__ xy,               // (B, N, 2)
    __hip_bfloat16* __restrict__ dsrc,        // (B, N, C)
    int B, int N, int C, int H, int W) {
  const long total = (long)B * N * C;
  const long HW = (long)H * W;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int c = i % C;
    long bn = i / C;
    int n = bn % N;
    long b = bn / N;
    int x = xy[bn * 2 + 0];
    int y = xy[bn * 2 + 1];
    dsrc[i] = dout[(b * C + c) * HW + (long)y * W + x];
  }
}
