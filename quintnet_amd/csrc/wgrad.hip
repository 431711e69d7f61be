// Hand-written CDNA4 (gfx950) bf16 TN weight-gradient GEMM with split-K.
//
//   dW[N,K] = dY[M,N]^T · X[M,K]      (nn.Linear wgrad; contraction over M)
//
// hipBLASLt runs the bench's wgrad shapes at only 208-513 TF (the output is
// small — 768..3072 per side — and the contraction is 16384 deep), so this
// kernel split-Ks the M axis across blocks: each (n-tile, k-tile, split)
// block accumulates a 128x128 fp32 tile over its M chunk into a per-split
// slab; a tiny fold kernel sums the slabs into the bf16 result
// (deterministic — no atomics).
//
// Both MFMA operands need fragments ALONG M, which is the strided dim of
// both inputs, so tiles are staged with `global_load_lds` in a blocked
// [m/4][o/16][4][16] image and fragments are read with the gfx950 transpose
// instruction `ds_read_b64_tr_b16`: each 16-lane group reads one contiguous
// 64-element block as a row-major [4 m][16 o] matrix and lane i receives
// column i — i.e. 4 contract elements for out-row/col i (mapping verified
// empirically, tools/probe_tr.hip).  Two tr-reads build one 8-deep MFMA
// fragment.  Double-buffered LDS, one barrier per M-step (the proven
// structure of gemm.hip's 128² kernel).
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8w;
typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 bf16x4w;

__device__ __forceinline__ void wglds16(const void* g, void* lds) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)g,
      (__attribute__((address_space(3))) unsigned int*)(uintptr_t)(lds), 16, 0, 0);
}

__device__ __forceinline__ bf16x4w tr16(const unsigned short* lds) {
  return __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (__attribute__((address_space(3))) bf16x4w*)(uintptr_t)lds);
}

// image: per operand 128x64 (out-dim x m) = 128 blocks of 128B; block
// B = o16*16 + m4 at element offset B*64 holds rows m4*4..+3, cols o16*16..+15
// (row-major [4][16]).  A glds instr I fills blocks I*8..I*8+7: lane L ->
// block I*8+(L>>3), granule gb=L&7 -> (m = +gb>>1, o-half = gb&1).
#define WG_IMG_EL 8192  // elements per operand image (16 KiB)

__global__ __launch_bounds__(256) void wgrad_tn_kernel(
    const unsigned short* __restrict__ dY,  // [M,N]
    const unsigned short* __restrict__ X,   // [M,K]
    float* __restrict__ slab,               // [splits, N, K]
    int M, int N, int K, int chunk) {
  __shared__ unsigned short smem[2 * 2 * WG_IMG_EL];

  const int nbk = K >> 7;
  const int bn = blockIdx.x / nbk, bk = blockIdx.x % nbk;
  const int n0 = bn << 7, k0 = bk << 7;
  const long long m0 = (long long)blockIdx.y * chunk;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wr = wave >> 1, wc = wave & 1;  // 2x2 waves; per-wave 64n x 64k

  // glds source pointers: 4 instrs per operand per wave, bumped by m-step
  const unsigned short* gpa[4];
  const unsigned short* gpb[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    int I = wave * 4 + j;
    int Bl = I * 8 + (lane >> 3);
    int m4 = Bl & 15, o16 = Bl >> 4;
    int gb = lane & 7;
    long long mrow = m0 + m4 * 4 + (gb >> 1);
    gpa[j] = dY + mrow * N + n0 + o16 * 16 + (gb & 1) * 8;
    gpb[j] = X + mrow * K + k0 + o16 * 16 + (gb & 1) * 8;
  }

#define WG_STAGE(buf, sa, sb)                                                  \
  do {                                                                         \
    _Pragma("unroll") for (int j = 0; j < 4; ++j) {                            \
      int I = wave * 4 + j;                                                    \
      wglds16(gpa[j] + (sa), smem + (buf) * 2 * WG_IMG_EL + I * 512);          \
      wglds16(gpb[j] + (sb), smem + (buf) * 2 * WG_IMG_EL + WG_IMG_EL + I * 512); \
    }                                                                          \
  } while (0)

  // fragment tr-read element offsets: lane (frow = out row/col 0..15,
  // kq = m-quarter 0..3); tile f adds an o16 group; ks adds 8 m (512 el),
  // the second tr of a fragment adds 4 m (64 el).
  const int frow = lane & 15;
  const int kq = lane >> 4;
  const int abase = (wr * 4) * 16 * 64 + kq * 128 + frow * 4;
  const int bbase = WG_IMG_EL + (wc * 4) * 16 * 64 + kq * 128 + frow * 4;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  union frag8 { bf16x8w v; bf16x4w h[2]; };

#define WG_MFMA(buf)                                                           \
  do {                                                                         \
    const unsigned short* base = smem + (buf) * 2 * WG_IMG_EL;                 \
    _Pragma("unroll") for (int ks = 0; ks < 2; ++ks) {                         \
      frag8 af[4], bf[4];                                                      \
      _Pragma("unroll") for (int f = 0; f < 4; ++f) {                          \
        af[f].h[0] = tr16(base + abase + f * 1024 + ks * 512);                 \
        af[f].h[1] = tr16(base + abase + f * 1024 + ks * 512 + 64);            \
        bf[f].h[0] = tr16(base + bbase + f * 1024 + ks * 512);                 \
        bf[f].h[1] = tr16(base + bbase + f * 1024 + ks * 512 + 64);            \
      }                                                                        \
      _Pragma("unroll") for (int i = 0; i < 4; ++i)                            \
          _Pragma("unroll") for (int j = 0; j < 4; ++j)                        \
              acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(             \
                  af[i].v, bf[j].v, acc[i][j], 0, 0, 0);                       \
    }                                                                          \
  } while (0)

  // ---- double-buffered pipelined loop (raw barriers + counted vmcnt) ------
  // glds for tile s+2 is issued AFTER the barrier that retires tile s's
  // reads, and stays in flight across both barriers of step s+1 (guide:
  // 2-buffer overlap, +40% over the serial __syncthreads pattern whose
  // fence drains the glds queue).  vmcnt(8) = one tile (8 glds/wave) in
  // flight; every wave waits its own count BEFORE the barrier, so data
  // staged by other waves is landed block-wide when the MFMAs read it.
  const int nsteps = chunk >> 6;
  const long long da = (long long)64 * N, db = (long long)64 * K;
  WG_STAGE(0, 0, 0);
  if (nsteps > 1) WG_STAGE(1, da, db);
  long long sa = da, sb = db;
  for (int s = 0; s < nsteps - 1; ++s) {
    asm volatile("s_waitcnt vmcnt(8)" ::: "memory");  // tile s landed; s+1 flying
    __builtin_amdgcn_s_barrier();
    WG_MFMA(s & 1);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    if (s + 2 < nsteps) {
      sa += da; sb += db;
      WG_STAGE(s & 1, sa, sb);
    }
  }
  // last step: nothing is staged behind it, so its own tile is the youngest
  // in flight — drain fully.
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  WG_MFMA((nsteps - 1) & 1);

  // ---- epilogue: fp32 partial tile into this split's slab -----------------
  const int erow = (lane >> 4) * 4;
  const int ecol = lane & 15;
  float* out = slab + (long long)blockIdx.y * N * K;
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int n = n0 + wr * 64 + i * 16 + erow;
      const int k = k0 + wc * 64 + j * 16 + ecol;
#pragma unroll
      for (int r = 0; r < 4; ++r)
        out[(long long)(n + r) * K + k] = acc[i][j][r];
    }
#undef WG_MFMA
#undef WG_STAGE
}

// 32-m-row stage variant: halves the LDS footprint (2x2x8 KiB = 32 KiB)
// so the CU fits 4 workgroups instead of 2 — 3 waves/SIMD after the
// VGPR cap instead of the 2 the 64 KiB default allows (the default
// kernel is LDS-limited: profiles/kernel_resources.md).  Twice the
// steps, half the MFMA work per step, same counted-vmcnt two-buffer
// pipeline.  Opt-in via QN_WGRAD_LDS=32 until A/B'd on hardware (r3).
#define WG32_IMG_EL 4096  // elements per operand image (8 KiB)

__global__ __launch_bounds__(256, 3) void wgrad_tn32_kernel(
    const unsigned short* __restrict__ dY,  // [M,N]
    const unsigned short* __restrict__ X,   // [M,K]
    float* __restrict__ slab,               // [splits, N, K]
    int M, int N, int K, int chunk) {
  __shared__ unsigned short smem[2 * 2 * WG32_IMG_EL];

  const int nbk = K >> 7;
  const int bn = blockIdx.x / nbk, bk = blockIdx.x % nbk;
  const int n0 = bn << 7, k0 = bk << 7;
  const long long m0 = (long long)blockIdx.y * chunk;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wr = wave >> 1, wc = wave & 1;  // 2x2 waves; per-wave 64n x 64k

  // image per operand: 128 out x 32 m = 64 blocks (B = o16*8 + m4,
  // m4 0..7); 8 glds instrs fill it -> 2 per wave per operand
  const unsigned short* gpa[2];
  const unsigned short* gpb[2];
#pragma unroll
  for (int j = 0; j < 2; ++j) {
    int I = wave * 2 + j;
    int Bl = I * 8 + (lane >> 3);
    int m4 = Bl & 7, o16 = Bl >> 3;
    int gb = lane & 7;
    long long mrow = m0 + m4 * 4 + (gb >> 1);
    gpa[j] = dY + mrow * N + n0 + o16 * 16 + (gb & 1) * 8;
    gpb[j] = X + mrow * K + k0 + o16 * 16 + (gb & 1) * 8;
  }

#define WG32_STAGE(buf, sa, sb)                                                  do {                                                                             _Pragma("unroll") for (int j = 0; j < 2; ++j) {                                  int I = wave * 2 + j;                                                          wglds16(gpa[j] + (sa), smem + (buf) * 2 * WG32_IMG_EL + I * 512);              wglds16(gpb[j] + (sb), smem + (buf) * 2 * WG32_IMG_EL + WG32_IMG_EL + I * 512);     }                                                                            } while (0)

  // fragment offsets: o16 group stride is now 8 blocks x 64 = 512 el;
  // single 32-m depth per step (the ks loop of the 64-m kernel is gone)
  const int frow = lane & 15;
  const int kq = lane >> 4;
  const int abase = (wr * 4) * 8 * 64 + kq * 128 + frow * 4;
  const int bbase = WG32_IMG_EL + (wc * 4) * 8 * 64 + kq * 128 + frow * 4;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  union frag8 { bf16x8w v; bf16x4w h[2]; };

#define WG32_MFMA(buf)                                                           do {                                                                             const unsigned short* base = smem + (buf) * 2 * WG32_IMG_EL;                   frag8 af[4], bf[4];                                                            _Pragma("unroll") for (int f = 0; f < 4; ++f) {                                  af[f].h[0] = tr16(base + abase + f * 512);                                     af[f].h[1] = tr16(base + abase + f * 512 + 64);                                bf[f].h[0] = tr16(base + bbase + f * 512);                                     bf[f].h[1] = tr16(base + bbase + f * 512 + 64);                              }                                                                              _Pragma("unroll") for (int i = 0; i < 4; ++i)                                      _Pragma("unroll") for (int j = 0; j < 4; ++j)                                      acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(                               af[i].v, bf[j].v, acc[i][j], 0, 0, 0);                           } while (0)

  const int nsteps = chunk >> 5;
  const long long da = (long long)32 * N, db = (long long)32 * K;
  WG32_STAGE(0, 0, 0);
  if (nsteps > 1) WG32_STAGE(1, da, db);
  long long sa = da, sb = db;
  for (int s = 0; s < nsteps - 1; ++s) {
    // 4 glds/wave per tile: tile s landed once the s+1 tile's 4 are the
    // only ones outstanding
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    WG32_MFMA(s & 1);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    if (s + 2 < nsteps) {
      sa += da; sb += db;
      WG32_STAGE(s & 1, sa, sb);
    }
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  WG32_MFMA((nsteps - 1) & 1);

  const int erow = (lane >> 4) * 4;
  const int ecol = lane & 15;
  float* out = slab + (long long)blockIdx.y * N * K;
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int n = n0 + wr * 64 + i * 16 + erow;
      const int k = k0 + wc * 64 + j * 16 + ecol;
#pragma unroll
      for (int r = 0; r < 4; ++r)
        out[(long long)(n + r) * K + k] = acc[i][j][r];
    }
#undef WG32_MFMA
#undef WG32_STAGE
}

// fold: dW_bf16[i] = sum_s slab[s][i]
__global__ void wgrad_fold_kernel(const float* __restrict__ slab,
                                  unsigned short* __restrict__ out,
                                  long long nk, int splits) {
  const long long i0 = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  if (i0 >= nk) return;
  float a[4] = {0.f, 0.f, 0.f, 0.f};
  for (int s = 0; s < splits; ++s) {
    const float* p = slab + (long long)s * nk + i0;
#pragma unroll
    for (int j = 0; j < 4; ++j) a[j] += p[j];
  }
#pragma unroll
  for (int j = 0; j < 4; ++j)
    if (i0 + j < nk) out[i0 + j] = f32_to_bf16(a[j]);
}

#include <cstdlib>

int wgrad_tn_splits(int M, int N, int K) {
  static int force = -2;
  if (force == -2) {
    const char* e = getenv("QN_WGRAD_SPLITS");
    force = e ? atoi(e) : -1;
  }
  const long long ntiles = (long long)(N >> 7) * (K >> 7);
  if (force > 0 && M % (64 * force) == 0) return force;
  // cap tuned by sweep (tools/sweep_wgrad.py, profiles r04): 108 tiles -> 4,
  // 36 -> 8; the >=144-tile shapes go to the library anyway (linear.py).
  int splits = 1;
  while (splits < 32 && ntiles * splits * 2 <= 512 &&
         M % (64 * splits * 2) == 0)
    splits *= 2;
  return splits;
}

void wgrad_tn_launch(const unsigned short* dY, const unsigned short* X,
                     float* slab, unsigned short* out, int M, int N, int K,
                     int splits, hipStream_t stream) {
  const int chunk = M / splits;
  dim3 grid((N >> 7) * (K >> 7), splits), blk(256);
  static int lds32 = -1;
  if (lds32 < 0) {
    const char* e = getenv("QN_WGRAD_LDS");
    lds32 = (e && atoi(e) == 32) ? 1 : 0;
  }
  if (lds32 && (chunk & 31) == 0)
    hipLaunchKernelGGL(wgrad_tn32_kernel, grid, blk, 0, stream, dY, X, slab,
                       M, N, K, chunk);
  else
    hipLaunchKernelGGL(wgrad_tn_kernel, grid, blk, 0, stream, dY, X, slab, M,
                       N, K, chunk);
  const long long nk = (long long)N * K;
  hipLaunchKernelGGL(wgrad_fold_kernel, dim3((nk / 4 + 255) / 256), dim3(256),
                     0, stream, slab, out, nk, splits);
}
