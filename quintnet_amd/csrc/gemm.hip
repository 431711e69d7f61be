// Hand-written CDNA4 (gfx950) bf16 MFMA GEMM — the TP-linear forward path.
//
//   C[M,N] = act(A[M,K] · B[N,K]^T + bias)        (nn.Linear layout, NT)
//
// Design (guide: cdna_hip_programming.md §5):
//  * 128×128 block tile, BK=64; 4 waves (2×2), each computing a 64×64
//    sub-tile as 4×4 fragments of v_mfma_f32_16x16x32_bf16 (fp32 accum).
//  * DOUBLE-BUFFERED LDS with ONE barrier per K-tile: next tile stages
//    into the other buffer while the current one feeds the MFMAs.
//  * interior tiles stage via async `global_load_lds` (dwordx4, 16 B per
//    lane — guide Common-mistake #1: width 16 is the 1.7× lever); the
//    LDS image is lane-linear, so the T2 XOR bank-swizzle moves to the
//    per-lane SOURCE address (guide §5.4 rule 21) with the matching XOR
//    on the fragment reads;
//  * edge tiles (M/N/K remainders) fall back to bounds-checked
//    register staging producing the SAME LDS image (T14 split: loads
//    issue before the MFMA cluster, ds_writes after);
//  * bias add + GELU/ReLU fused into the epilogue (optionally emitting
//    the pre-activation for backward);
//  * XCD-aware bijective blockIdx swizzle (T1) for L2 affinity.
//
// Replaces the implicit cuBLAS GEMMs of reference tensor_parallel/layers.py
// :119,:211 and utils/GPT2 (c_attn/c_proj/c_fc/lm_head).
#include "common.h"

#define BM 128
#define BN 128
#define BK 64
#define QN_ACT_NONE 0
#define QN_ACT_GELU 1
#define QN_ACT_RELU 2

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

__device__ __forceinline__ s16x8 zero8() {
  s16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
  return z;
}

// async 16B global->LDS (lane-linear dest). lds generic pointer cast per
// the CK idiom; gfx950 supports 16 B per lane.
__device__ __forceinline__ void glds16(const void* g, void* lds) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)g,
      (__attribute__((address_space(3))) unsigned int*)(uintptr_t)(lds), 16, 0, 0);
}

// ---------------------------------------------------------------------------
// Staging.  LDS image (both paths identical): linear [128][64] tile where
// LDS(row, g) holds GLOBAL column group g ^ (row & 7)  (g = 16B group 0..7).
// Fragment reads therefore XOR their group index with (row & 7).
// ---------------------------------------------------------------------------
// fast path: one glds instruction stages 8 rows (64 lanes x 16B);
// NWAVES waves cover ROWS rows with ROWS/(8*NWAVES) calls each.  The
// per-lane source pointers are precomputed once and bumped by BK per
// tile — the 64-bit address rebuild per tile was ~40% of the kernel's
// VALU issue (PMC: 6.2 VALU/MFMA).
template <int ROWS, int NWAVES, int CALLS = ROWS / (8 * NWAVES)>
__device__ __forceinline__ void stage_glds_pre(
    unsigned short* lds, const unsigned short* const (&gp)[CALLS], int k_elems,
    int wave) {
#pragma unroll
  for (int j = 0; j < CALLS; ++j)
    glds16(gp[j] + k_elems, lds + (wave * (ROWS / NWAVES) + j * 8) * BK);
}

// slow path (edges): bounds-checked loads to regs ...
template <int NTHREADS, int PASSES>
__device__ __forceinline__ void load_tile_regs(
    const unsigned short* __restrict__ src, long long ld, int row0, int rows,
    int k0, int K, s16x8 (&regs)[PASSES]) {
#pragma unroll
  for (int p = 0; p < PASSES; ++p) {
    int lin = p * NTHREADS + threadIdx.x;
    int row = lin >> 3;
    int g = (lin & 7) ^ (row & 7);  // read the swizzled source group
    int k = k0 + g * 8;
    bool ok = (row0 + row < rows) && (k < K);
    regs[p] = ok ? *reinterpret_cast<const s16x8*>(src + (long long)(row0 + row) * ld + k)
                 : zero8();
  }
}

// ... then linear ds_writes (same image as glds)
template <int NTHREADS, int PASSES>
__device__ __forceinline__ void write_tile_lds(unsigned short* lds, const s16x8 (&regs)[PASSES]) {
#pragma unroll
  for (int p = 0; p < PASSES; ++p) {
    int lin = p * NTHREADS + threadIdx.x;
    int row = lin >> 3;
    int g = lin & 7;
    *reinterpret_cast<s16x8*>(lds + row * BK + g * 8) = regs[p];
  }
}

template <int ACT, bool SAVE_PRE, bool USE_GLDS, int TBM, int TBN, int WR, int WC,
          int MINW = 0>
__global__ __launch_bounds__(WR * WC * 64, MINW) void gemm_nt_kernel(
    const unsigned short* __restrict__ A,  // [M,K]
    const unsigned short* __restrict__ B,  // [N,K]
    const unsigned short* __restrict__ bias,  // [N] or nullptr
    unsigned short* __restrict__ C,        // [M,N]
    unsigned short* __restrict__ Cpre,     // [M,N] pre-activation (SAVE_PRE)
    int M, int N, int K) {
  constexpr int NWAVES = WR * WC;
  constexpr int NTHREADS = NWAVES * 64;
  // two double-buffered tiles: buffer b: A at b*(TBM+TBN)*BK, B after A
  __shared__ unsigned short smem[2 * (TBM + TBN) * BK];
#define As(b) (smem + (b) * (TBM + TBN) * BK)
#define Bs(b) (smem + (b) * (TBM + TBN) * BK + TBM * BK)

  // XCD-aware bijective block swizzle (T1)
  const int nbm = (M + TBM - 1) / TBM;
  const int nbn = (N + TBN - 1) / TBN;
  const int nwg = nbm * nbn;
  int bid = blockIdx.x;
  if (nwg >= 16) {
    const int nx = 8;
    int qq = nwg / nx, rr = nwg % nx;
    int xcd = bid % nx, idx = bid / nx;
    bid = (xcd < rr ? xcd * (qq + 1) : rr * (qq + 1) + (xcd - rr) * qq) + idx;
  }
  const int bm = bid / nbn;
  const int bn = bid % nbn;
  const int m0 = bm * TBM;
  const int n0 = bn * TBN;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wr = wave / WC;
  const int wc = wave % WC;

  const bool interior_mn = (m0 + TBM <= M) && (n0 + TBN <= N);

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int frow = lane & 15;       // fragment row/col within 16
  const int kq = lane >> 4;         // k quarter (0..3) of the 32-K step

  // precomputed per-lane glds source pointers (k advances via offset)
  constexpr int ACALLS = TBM / (8 * NWAVES);
  constexpr int BCALLS = TBN / (8 * NWAVES);
  const unsigned short* gpa[ACALLS];
  const unsigned short* gpb[BCALLS];
#pragma unroll
  for (int j = 0; j < ACALLS; ++j) {
    int row = wave * (TBM / NWAVES) + j * 8 + (lane >> 3);
    int g = (lane & 7) ^ (row & 7);  // source-side swizzle (rule 21)
    gpa[j] = A + (long long)(m0 + row) * K + g * 8;
  }
#pragma unroll
  for (int j = 0; j < BCALLS; ++j) {
    int row = wave * (TBN / NWAVES) + j * 8 + (lane >> 3);
    int g = (lane & 7) ^ (row & 7);
    gpb[j] = B + (long long)(n0 + row) * K + g * 8;
  }
  // precomputed LDS fragment offsets (elements); ks=1 toggles bit 5 (^32)
  int aoff[4], boff[4];
#pragma unroll
  for (int f = 0; f < 4; ++f) {
    int arow = wr * 64 + f * 16 + frow;
    aoff[f] = arow * BK + ((kq ^ (arow & 7)) << 3);
    int brow = wc * 64 + f * 16 + frow;
    boff[f] = brow * BK + ((kq ^ (brow & 7)) << 3);
  }

#define QN_MFMA_TILE(at, bt)                                                     _Pragma("unroll")                                                              for (int ks = 0; ks < 2; ++ks) {                                                 bf16x8 af[4], bf[4];                                                           _Pragma("unroll")                                                              for (int f = 0; f < 4; ++f) {                                                    af[f] = *reinterpret_cast<const bf16x8*>(&(at)[aoff[f] ^ (ks << 5)]);          bf[f] = *reinterpret_cast<const bf16x8*>(&(bt)[boff[f] ^ (ks << 5)]);        }                                                                              _Pragma("unroll")                                                              for (int i = 0; i < 4; ++i)                                                      _Pragma("unroll")                                                              for (int j = 0; j < 4; ++j)                                                      acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(                               af[i], bf[j], acc[i][j], 0, 0, 0);                                   }

  // ---- prologue: stage tile 0 into buffer 0 -------------------------------
  constexpr int APASS = TBM * 8 / NTHREADS;
  constexpr int BPASS = TBN * 8 / NTHREADS;
  s16x8 ra[APASS], rb[BPASS];
  const bool all_fast = USE_GLDS && interior_mn && (K % BK == 0);
  if (all_fast) {
    // fully-interior fast loop: unrolled x2 so buffer pointers and LDS
    // offsets stay loop-invariant; glds sources advance by += k
    stage_glds_pre<TBM, NWAVES>(As(0), gpa, 0, wave);
    stage_glds_pre<TBN, NWAVES>(Bs(0), gpb, 0, wave);
    int k0 = 0;
    while (true) {
      // even tile in buf0
      __syncthreads();
      if (k0 + BK < K) {
        stage_glds_pre<TBM, NWAVES>(As(1), gpa, k0 + BK, wave);
        stage_glds_pre<TBN, NWAVES>(Bs(1), gpb, k0 + BK, wave);
      }
      QN_MFMA_TILE(As(0), Bs(0));
      k0 += BK;
      if (k0 >= K) break;
      // odd tile in buf1
      __syncthreads();
      if (k0 + BK < K) {
        stage_glds_pre<TBM, NWAVES>(As(0), gpa, k0 + BK, wave);
        stage_glds_pre<TBN, NWAVES>(Bs(0), gpb, k0 + BK, wave);
      }
      QN_MFMA_TILE(As(1), Bs(1));
      k0 += BK;
      if (k0 >= K) break;
    }
  } else {
    bool fast0 = USE_GLDS && interior_mn && (BK <= K);
    if (fast0) {
      stage_glds_pre<TBM, NWAVES>(As(0), gpa, 0, wave);
      stage_glds_pre<TBN, NWAVES>(Bs(0), gpb, 0, wave);
    } else {
      load_tile_regs<NTHREADS, APASS>(A, K, m0, M, 0, K, ra);
      load_tile_regs<NTHREADS, BPASS>(B, K, n0, N, 0, K, rb);
      write_tile_lds<NTHREADS, APASS>(As(0), ra);
      write_tile_lds<NTHREADS, BPASS>(Bs(0), rb);
    }
    int cur = 0;
    for (int k0 = 0; k0 < K; k0 += BK) {
      __syncthreads();  // buf[cur] complete (drains in-flight glds too)
      const int kn = k0 + BK;
      const bool have_next = kn < K;
      const bool fast_next = USE_GLDS && interior_mn && (kn + BK <= K);
      if (have_next) {
        if (fast_next) {
          stage_glds_pre<TBM, NWAVES>(As(cur ^ 1), gpa, kn, wave);
          stage_glds_pre<TBN, NWAVES>(Bs(cur ^ 1), gpb, kn, wave);
        } else {
          load_tile_regs<NTHREADS, APASS>(A, K, m0, M, kn, K, ra);  // issue now,
          load_tile_regs<NTHREADS, BPASS>(B, K, n0, N, kn, K, rb);  // write later (T14)
        }
      }
      const unsigned short* at = As(cur);
      const unsigned short* bt = Bs(cur);
      QN_MFMA_TILE(at, bt);
      if (have_next && !fast_next) {
        write_tile_lds<NTHREADS, APASS>(As(cur ^ 1), ra);
        write_tile_lds<NTHREADS, BPASS>(Bs(cur ^ 1), rb);
      }
      cur ^= 1;
    }
  }
#undef QN_MFMA_TILE

  // epilogue: D[i][j] lane map col = lane&15, row = (lane>>4)*4 + reg
  const int erow = (lane >> 4) * 4;
  const int ecol = lane & 15;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int col = n0 + wc * 64 + j * 16 + ecol;
      if (col >= N) continue;
      float bv = (bias != nullptr) ? bf16_to_f32(bias[col]) : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + wr * 64 + i * 16 + erow + r;
        if (row >= M) continue;
        float v = acc[i][j][r] + bv;
        long long idx = (long long)row * N + col;
        if constexpr (SAVE_PRE) Cpre[idx] = f32_to_bf16(v);
        if constexpr (ACT == QN_ACT_GELU) v = gelu_tanh(v);
        else if constexpr (ACT == QN_ACT_RELU) v = fmaxf(v, 0.f);
        C[idx] = f32_to_bf16(v);
      }
    }
  }
}

// ===========================================================================
// 256x256 8-phase deep-pipelined kernel (guide §"The 256² 8-phase template"):
// 8 waves (2M x 4N), per-wave C = 128x64 (128 acc VGPRs); the K-step (BK=64)
// is split into 4 quadrant phases of 16 MFMAs each; fragment ds_reads run one
// phase ahead and BOTH A quadrant-sets and B quadrant-sets stay register-
// resident, so every LDS region is read exactly once per K-tile.  Staging is
// glds-only (2 x global_load_lds_dwordx4 per wave per phase = one "half-tile"),
// with RAW s_barrier + lgkmcnt(0) (never __syncthreads: its fence would drain
// the in-flight glds queue, the ~20% stall of the 2-barrier structure) and a
// counted s_waitcnt vmcnt(6) only at the two K-tile boundaries (3 half-tiles
// stay in flight across barriers).  LDS = 2 buffers x (A+B) 256x64 = 128 KiB,
// st_16x32 XOR-swizzled via the per-lane SOURCE address (the glds LDS dest is
// lane-linear).  One block/CU, 512 threads, ~250 VGPRs.
//
// Derived half-tile schedule (phase -> {MFMA quadrant | prep reads | stage}):
//   p0: Q00(T) | BO(T)->B1        | AE(T+2)        Q00=(QM0,QN0) uses A0,B0
//   p1: Q01(T) | AO(T)->A1        | BE(T+2)        Q01 uses A0,B1
//   p2: Q10(T) | -                | BO(T+2)  [vmcnt(6) before trailing barrier]
//   p3: Q11(T) | AE,BE(T+1)->A0,B0| AO(T+2)
//   p4..p7: same over tile T+1, staging T+3.
// Landing/overwrite safety for every phase was checked against the retire
// points (each region's single read) — see the schedule table in the commit.
// ===========================================================================

#define P8_LDSEL (2 * 512 * 64)  // elements: 2 buf x (256 A + 256 B) x 64

// per-lane source column group for a glds whose 64 lanes fill one 8-row x
// 128B LDS subtile: st_16x32 swizzle = flip granule bit1 when row bit2 set
// (row bit2 == lane bit5 within a subtile)
__device__ __forceinline__ int p8_src_koct(int lane) {
  return (lane & 7) ^ (((lane >> 5) & 1) << 1);
}

template <int ACT, bool SAVE_PRE>
__global__ __launch_bounds__(512, 1) void gemm_nt_8p_kernel(
    const unsigned short* __restrict__ A,  // [M,K]
    const unsigned short* __restrict__ B,  // [N,K]
    const unsigned short* __restrict__ bias,
    unsigned short* __restrict__ C, unsigned short* __restrict__ Cpre,
    int M, int N, int K) {
  __shared__ unsigned short smem[P8_LDSEL];

  const int nbm = M >> 8, nbn = N >> 8;
  const int nwg = nbm * nbn;
  int bid = blockIdx.x;
  if (nwg >= 16) {
    int qq = nwg / 8, rr = nwg % 8;
    int xcd = bid % 8, idx = bid / 8;
    bid = (xcd < rr ? xcd * (qq + 1) : rr * (qq + 1) + (xcd - rr) * qq) + idx;
  }
  const int m0 = (bid / nbn) << 8;
  const int n0 = (bid % nbn) << 8;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wr = wave >> 2;  // 0..1 (M)
  const int wc = wave & 3;   // 0..3 (N)

  // ---- per-lane glds source pointers + wave-uniform LDS dests -------------
  // half-tiles: 0=AE (A rows {0-63,128-191}), 1=BE (B rows {64wc..+32}),
  //             2=BO (BE+32), 3=AO (AE+64); each = 16 subtiles of 8 rows,
  //             wave w stages subtiles 2w, 2w+1.
  // wave w stages subtiles 2w and 2w+1 of each half; the pair is always 8
  // rows apart (pairs never straddle a stripe boundary), so only the first
  // subtile's pointer is kept (+8*K / +512 for the second) — 8 fewer VGPRs,
  // which is what keeps this kernel spill-free at the 256-VGPR cap.
  const unsigned short* gsrc[4];
  int ldst[4];
  const int skoct = p8_src_koct(lane);
  const int srow = lane >> 3;
  {
    int st = 2 * wave;
    int ra = ((st & 8) << 4) + ((st & 7) << 3);        // AE subtile row0
    int rb = ((st >> 2) << 6) + ((st & 3) << 3);       // BE subtile row0
    gsrc[0] = A + (long long)(m0 + ra + srow) * K + skoct * 8;
    gsrc[3] = A + (long long)(m0 + ra + 64 + srow) * K + skoct * 8;
    gsrc[1] = B + (long long)(n0 + rb + srow) * K + skoct * 8;
    gsrc[2] = B + (long long)(n0 + rb + 32 + srow) * K + skoct * 8;
    ldst[0] = ra * 64;
    ldst[3] = (ra + 64) * 64;
    ldst[1] = (256 + rb) * 64;
    ldst[2] = (256 + rb + 32) * 64;
  }
  const long long row8 = (long long)8 * K;

#define P8_STAGE(h, buf, kel)                                                  \
  do {                                                                         \
    glds16(gsrc[h] + (kel), smem + (buf) * (512 * 64) + ldst[h]);              \
    glds16(gsrc[h] + row8 + (kel), smem + (buf) * (512 * 64) + ldst[h] + 512); \
  } while (0)

  // ---- fragment LDS offsets (elements), ks=1 toggles +32 ------------------
  const int frow = lane & 15;
  const int kq = lane >> 4;
  int aoffE[4], boffE[2];
#pragma unroll
  for (int f = 0; f < 4; ++f) {
    int row = wr * 128 + f * 16 + frow;
    aoffE[f] = row * 64 + ((kq ^ (((row >> 2) & 1) << 1)) << 3);
  }
#pragma unroll
  for (int f = 0; f < 2; ++f) {
    int row = wc * 64 + f * 16 + frow;
    boffE[f] = (256 + row) * 64 + ((kq ^ (((row >> 2) & 1) << 1)) << 3);
  }

  bf16x8 Ar[2][8];  // [QM set][f*2+ks]
  bf16x8 Br[2][4];  // [QN set][f*2+ks]
  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // extra: AO rows are +64 (+4096 el), BO rows +32 (+2048 el); row bit2 —
  // and so the swizzle XOR — is unchanged by either shift.
#define P8_READ_A(set, buf, extra)                                            \
  _Pragma("unroll") for (int f = 0; f < 4; ++f) _Pragma("unroll")             \
      for (int ks = 0; ks < 2; ++ks)                                          \
          Ar[set][f * 2 + ks] = *reinterpret_cast<const bf16x8*>(             \
              &smem[(buf) * (512 * 64) + ((aoffE[f] + (extra)) ^ (ks << 5))]);
#define P8_READ_B(set, buf, extra)                                            \
  _Pragma("unroll") for (int f = 0; f < 2; ++f) _Pragma("unroll")             \
      for (int ks = 0; ks < 2; ++ks)                                          \
          Br[set][f * 2 + ks] = *reinterpret_cast<const bf16x8*>(             \
              &smem[(buf) * (512 * 64) + ((boffE[f] + (extra)) ^ (ks << 5))]);

#define P8_MFMA(QM, QN)                                                       \
  _Pragma("unroll") for (int ks = 0; ks < 2; ++ks) _Pragma("unroll")          \
      for (int i = 0; i < 4; ++i) _Pragma("unroll") for (int j = 0; j < 2;    \
                                                         ++j)                 \
          acc[(QM) * 4 + i][(QN) * 2 + j] =                                   \
              __builtin_amdgcn_mfma_f32_16x16x32_bf16(                        \
                  Ar[QM][i * 2 + ks], Br[QN][j * 2 + ks],                     \
                  acc[(QM) * 4 + i][(QN) * 2 + j], 0, 0, 0);

#define P8_BAR() __builtin_amdgcn_s_barrier()
#define P8_LGKM0() asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory")
#define P8_VM(n) asm volatile("s_waitcnt vmcnt(" #n ")" ::: "memory")

// one phase: prep reads + one half-tile stage, raw barrier, drained LDS,
// prioritized MFMA cluster; VMW expands to the counted vmcnt before the
// trailing barrier on the two boundary phases (cross-wave: every wave waits
// its own count BEFORE the barrier that precedes the dependent reads).
#define P8_PHASE(PREP, STAGE, QM, QN, VMW)                                    \
  PREP;                                                                       \
  STAGE;                                                                      \
  P8_BAR();                                                                   \
  P8_LGKM0();                                                                 \
  __builtin_amdgcn_s_setprio(1);                                              \
  P8_MFMA(QM, QN);                                                            \
  __builtin_amdgcn_s_setprio(0);                                              \
  VMW;                                                                        \
  P8_BAR();

  // ---- prologue: stage tiles 0 (buf0) and 1 (buf1) ------------------------
  P8_STAGE(0, 0, 0);
  P8_STAGE(1, 0, 0);
  P8_STAGE(2, 0, 0);
  P8_STAGE(3, 0, 0);
  P8_STAGE(0, 1, 64);
  P8_STAGE(1, 1, 64);
  P8_STAGE(2, 1, 64);
  P8_STAGE(3, 1, 64);
  P8_VM(8);  // tile 0 landed (per wave; barrier below makes it block-wide)
  P8_BAR();
  P8_READ_A(0, 0, 0);  // AE(0) -> A0
  P8_READ_B(0, 0, 0);  // BE(0) -> B0

  const int niter = K >> 7;  // K-tile pairs
  int k2 = 128, k3 = 192;    // k offsets (elements) of tiles T+2, T+3
  for (int it = 0; it < niter; ++it) {
    P8_PHASE(P8_READ_B(1, 0, 2048), P8_STAGE(0, 0, k2), 0, 0, );  // p0
    P8_PHASE(P8_READ_A(1, 0, 4096), P8_STAGE(1, 0, k2), 0, 1, );  // p1
    P8_PHASE(, P8_STAGE(2, 0, k2), 1, 0, P8_VM(6));               // p2
    P8_PHASE(P8_READ_A(0, 1, 0) P8_READ_B(0, 1, 0), P8_STAGE(3, 0, k2), 1, 1, );  // p3
    P8_PHASE(P8_READ_B(1, 1, 2048), P8_STAGE(0, 1, k3), 0, 0, );  // p4
    P8_PHASE(P8_READ_A(1, 1, 4096), P8_STAGE(1, 1, k3), 0, 1, );  // p5
    P8_PHASE(, P8_STAGE(2, 1, k3), 1, 0, P8_VM(6));               // p6
    P8_PHASE(P8_READ_A(0, 0, 0) P8_READ_B(0, 0, 0), P8_STAGE(3, 1, k3), 1, 1, );  // p7
    k2 += 128;
    if (k2 >= K) k2 -= K;  // tail stagings wrap to valid (unused) addresses
    k3 += 128;
    if (k3 >= K) k3 -= K;
  }

  // ---- epilogue: bias + activation, all-interior --------------------------
  const int erow = (lane >> 4) * 4;
  const int ecol = lane & 15;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const int col = n0 + wc * 64 + j * 16 + ecol;
    const float bv = (bias != nullptr) ? bf16_to_f32(bias[col]) : 0.f;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const long long base = (long long)(m0 + wr * 128 + i * 16 + erow) * N + col;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float v = acc[i][j][r] + bv;
        const long long idx = base + (long long)r * N;
        if constexpr (SAVE_PRE) Cpre[idx] = f32_to_bf16(v);
        if constexpr (ACT == QN_ACT_GELU) v = gelu_tanh(v);
        else if constexpr (ACT == QN_ACT_RELU) v = fmaxf(v, 0.f);
        C[idx] = f32_to_bf16(v);
      }
    }
  }
#undef P8_STAGE
}

// ===========================================================================
// PERSISTENT 8-phase kernel (the library-beating path for the skinny-K
// GPT-2 forward shapes, M=16384 K=768).  Same per-unit phase schedule as
// gemm_nt_8p_kernel, but each workgroup owns `cnt` whole 256² tiles and the
// glds pipeline runs CONTINUOUSLY across tile boundaries — the 2-K-tile
// prologue (the ~30-50% per-tile overhead at K=768) is paid once per WG
// instead of once per tile, and there is no inter-launch tail.
//
// Scheduling: with P=256 WGs (1/CU), workgroup w runs on XCD w%8 (HW
// dispatch affinity); tile_of(w,j) = (w%8)*32*cnt + j*32 + w/8 gives each
// XCD a CONTIGUOUS range of row-major tiles AND keeps its 32 concurrent
// WGs on 32 ADJACENT tiles in lockstep — at any instant an XCD's L2 only
// holds the ~400 KB of 64-deep A/B k-slabs those adjacent tiles share,
// so the nbn-fold A re-reads and nbm-fold B re-reads are L2 hits instead
// of HBM traffic.  Remainder tiles (T % 256) go to a second launch with
// P=T_rem, cnt=1 (tile offset t0).
//
// Epilogue stores between units mix with the counted vmcnt(6) boundary
// waits: vmcnt counts stores too on CDNA4, so those waits briefly also
// drain the C-tile stores (~sub-µs against an ~11 µs tile) — safe
// (over-waiting only), measured acceptable.
// ===========================================================================
template <int ACT, bool SAVE_PRE>
__global__ __launch_bounds__(512, 1) void gemm_nt_p8p_kernel(
    const unsigned short* __restrict__ A, const unsigned short* __restrict__ B,
    const unsigned short* __restrict__ bias, unsigned short* __restrict__ C,
    unsigned short* __restrict__ Cpre, int M, int N, int K, int nbn, int cnt,
    int t0) {
  __shared__ unsigned short smem[P8_LDSEL];
  const int w = blockIdx.x;
  const bool full = (gridDim.x == 256);

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wr = wave >> 2;
  const int wc = wave & 3;

  // source granule for the conflict-free image: LDS position (lane&7) of
  // row srow holds global granule (lane&7) ^ (srow<1:2>·{2,4})
  const int srow = lane >> 3;
  const int skoct =
      (lane & 7) ^ ((((srow >> 1) & 1) << 1) | (((srow >> 2) & 1) << 2));
  int ldst[4];
  int ra, rb;
  {
    int st = 2 * wave;
    ra = ((st & 8) << 4) + ((st & 7) << 3);
    rb = ((st >> 2) << 6) + ((st & 3) << 3);
    // wave-uniform LDS destinations — pin to SGPRs (as VGPRs they can
    // spill, and a scratch reload inside the loop carries a vmcnt(0)
    // that drains the whole staging pipeline)
    ldst[0] = __builtin_amdgcn_readfirstlane(ra * 64);
    ldst[3] = __builtin_amdgcn_readfirstlane((ra + 64) * 64);
    ldst[1] = __builtin_amdgcn_readfirstlane((256 + rb) * 64);
    ldst[2] = __builtin_amdgcn_readfirstlane((256 + rb + 32) * 64);
  }
  // staging via SRSRC buffer descriptors (T8): per-lane address work is a
  // single loop-CONSTANT 32-bit voffset; the tile/half/k-step components
  // are all scalar soffset math.  (A per-lane 64-bit pointer array here
  // spills at the 256-VGPR cap, and its scratch reloads inside the loop
  // each carry a vmcnt(0) that drains the glds pipeline.)
  const auto rsrcA = __builtin_amdgcn_make_buffer_rsrc(
      (void*)A, (short)0, (int)((long long)M * K * 2), 0x00020000);
  const auto rsrcB = __builtin_amdgcn_make_buffer_rsrc(
      (void*)B, (short)0, (int)((long long)N * K * 2), 0x00020000);
  const auto rsrcC = __builtin_amdgcn_make_buffer_rsrc(
      (void*)C, (short)0, (int)((long long)M * N * 2), 0x00020000);
  const auto rsrcP = __builtin_amdgcn_make_buffer_rsrc(
      (void*)(SAVE_PRE ? Cpre : C), (short)0, (int)((long long)M * N * 2),
      0x00020000);
  const int voffA = ((ra + srow) * K + skoct * 8) * 2;   // bytes, per lane
  const int voffB = ((rb + srow) * K + skoct * 8) * 2;
  const int dA1b = 64 * K * 2, dB1b = 32 * K * 2, row8b = 8 * K * 2;
  int soA = 0, soB = 0;  // per-tile scalar base offsets (bytes)
#define P8P_TILE(j) (full ? ((w & 7) * 32 * cnt + (j) * 32 + (w >> 3)) : (t0 + w))
#define P8P_SET_TILE(t)                                                        \
  do {                                                                         \
    soA = (((t) / nbn) << 8) * K * 2;                                          \
    soB = (((t) % nbn) << 8) * K * 2;                                          \
  } while (0)
#define P8_LDS(buf, h) ((__attribute__((address_space(3))) void*)(uintptr_t)( \
    smem + (buf) * (512 * 64) + ldst[h]))
#define P8_LDS2(buf, h) ((__attribute__((address_space(3))) void*)(uintptr_t)( \
    smem + (buf) * (512 * 64) + ldst[h] + 512))
  // half h: 0=AE  1=BE  2=BO(+dB1b)  3=AO(+dA1b); h is a literal → folds
#define P8_STAGE(h, buf, kel)                                                  \
  do {                                                                         \
    const int so_ = (((h) == 0 || (h) == 3) ? soA : soB) +                     \
                    ((h) == 3 ? dA1b : ((h) == 2 ? dB1b : 0)) + (kel) * 2;     \
    if ((h) == 0 || (h) == 3) {                                                \
      __builtin_amdgcn_raw_ptr_buffer_load_lds(rsrcA, P8_LDS(buf, h), 16,      \
                                               voffA, so_, 0, 0);              \
      __builtin_amdgcn_raw_ptr_buffer_load_lds(rsrcA, P8_LDS2(buf, h), 16,     \
                                               voffA, so_ + row8b, 0, 0);      \
    } else {                                                                   \
      __builtin_amdgcn_raw_ptr_buffer_load_lds(rsrcB, P8_LDS(buf, h), 16,      \
                                               voffB, so_, 0, 0);              \
      __builtin_amdgcn_raw_ptr_buffer_load_lds(rsrcB, P8_LDS2(buf, h), 16,     \
                                               voffB, so_ + row8b, 0, 0);      \
    }                                                                          \
  } while (0)

  // Fragment reads: one per-lane byte base per (operand, ks); buf/f/extra
  // fold into ds_read offset immediates.  The granule swizzle here is
  // CONFLICT-FREE for ds_read_b128's 4×16 lane groups: granule p stored
  // at position ((ks<<2)|kq) ^ (row<1:2>·{2,4}) — the old row-bit2-only
  // XOR measured a 50% SQ_LDS_BANK_CONFLICT rate (2-way); this one
  // simulates and measures clean.  Swizzle bit2 collides with the ks
  // offset, hence TWO bases per operand instead of a +64 add.
  const int frow = lane & 15;
  const int kq = lane >> 4;
  const char* smc = (const char*)smem;
  const int fr_ = ((((frow >> 1) & 1) << 1) | (((frow >> 2) & 1) << 2));
  int vA[2], vB[2];
#pragma unroll
  for (int ks = 0; ks < 2; ++ks) {
    const int p = ((ks << 2) | kq) ^ fr_;
    vA[ks] = (wr * 128 + frow) * 128 + p * 16;
    vB[ks] = (256 + wc * 64 + frow) * 128 + p * 16;
  }
#define P8P_READ_A(set, buf, extra)                                           \
  _Pragma("unroll") for (int f = 0; f < 4; ++f) _Pragma("unroll")             \
      for (int ks = 0; ks < 2; ++ks)                                          \
          Ar[set][f * 2 + ks] = *reinterpret_cast<const bf16x8*>(             \
              smc + (buf) * 65536 + vA[ks] + f * 2048 + (extra) * 2);
#define P8P_READ_B(set, buf, extra)                                           \
  _Pragma("unroll") for (int f = 0; f < 2; ++f) _Pragma("unroll")             \
      for (int ks = 0; ks < 2; ++ks)                                          \
          Br[set][f * 2 + ks] = *reinterpret_cast<const bf16x8*>(             \
              smc + (buf) * 65536 + vB[ks] + f * 2048 + (extra) * 2);

  bf16x8 Ar[2][8];
  bf16x8 Br[2][4];
  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int UT = K >> 7;          // 128-deep units per tile
  const int total = cnt * UT;     // units this WG
  P8P_SET_TILE(P8P_TILE(0));

  // prologue: stage unit 0 (K-tiles 0,1) into buf0,buf1
  P8_STAGE(0, 0, 0);
  P8_STAGE(1, 0, 0);
  P8_STAGE(2, 0, 0);
  P8_STAGE(3, 0, 0);
  P8_STAGE(0, 1, 64);
  P8_STAGE(1, 1, 64);
  P8_STAGE(2, 1, 64);
  P8_STAGE(3, 1, 64);
  P8_VM(8);
  P8_BAR();
  P8P_READ_A(0, 0, 0);
  P8P_READ_B(0, 0, 0);

  const int erow = (lane >> 4) * 4;
  const int ecol = lane & 15;

  // stage cursor = compute unit + 1, clamped at the chunk end; advanced
  // incrementally (an in-loop integer division spills at 256 VGPRs)
  int js = 0, kin_s = 0;
  if (total > 1) {
    kin_s = 1;
    if (kin_s == UT) {
      kin_s = 0;
      js = 1;
      P8P_SET_TILE(P8P_TILE(1));
    }
  }

  int cu = 0;
  for (int jc = 0; jc < cnt; ++jc) {
    for (int kin = 0; kin < UT; ++kin, ++cu) {
      const int k2 = kin_s << 7;
      const int k3 = k2 + 64;
      P8_PHASE(P8P_READ_B(1, 0, 2048), P8_STAGE(0, 0, k2), 0, 0, );
      P8_PHASE(P8P_READ_A(1, 0, 4096), P8_STAGE(1, 0, k2), 0, 1, );
      P8_PHASE(, P8_STAGE(2, 0, k2), 1, 0, P8_VM(6));
      P8_PHASE(P8P_READ_A(0, 1, 0) P8P_READ_B(0, 1, 0), P8_STAGE(3, 0, k2), 1, 1, );
      P8_PHASE(P8P_READ_B(1, 1, 2048), P8_STAGE(0, 1, k3), 0, 0, );
      P8_PHASE(P8P_READ_A(1, 1, 4096), P8_STAGE(1, 1, k3), 0, 1, );
      P8_PHASE(, P8_STAGE(2, 1, k3), 1, 0, P8_VM(6));
      P8_PHASE(P8P_READ_A(0, 0, 0) P8P_READ_B(0, 0, 0), P8_STAGE(3, 1, k3), 1, 1, );
      // advance the stage cursor unless already clamped at the end
      if (cu + 2 < total) {
        if (++kin_s == UT) {
          kin_s = 0;
          ++js;
          P8P_SET_TILE(P8P_TILE(js));
        }
      }
    }
    // tile complete: direct epilogue from acc (descriptor stores, 32-bit
    // byte offsets — 64-bit per-store address math spills), then reset
    {
      const int t = P8P_TILE(jc);
      const int m0 = (t / nbn) << 8;
      const int n0 = (t % nbn) << 8;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int col = n0 + wc * 64 + j * 16 + ecol;
        const float bv = (bias != nullptr) ? bf16_to_f32(bias[col]) : 0.f;
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          const int base = (m0 + wr * 128 + i * 16 + erow) * N + col;
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            float v = acc[i][j][r] + bv;
            const int off = (base + r * N) * 2;
            if constexpr (SAVE_PRE)
              __builtin_amdgcn_raw_buffer_store_b16(f32_to_bf16(v), rsrcP,
                                                    off, 0, 0);
            if constexpr (ACT == QN_ACT_GELU) v = gelu_tanh(v);
            else if constexpr (ACT == QN_ACT_RELU) v = fmaxf(v, 0.f);
            __builtin_amdgcn_raw_buffer_store_b16(f32_to_bf16(v), rsrcC, off,
                                                  0, 0);
          }
          acc[i][j] = {0.f, 0.f, 0.f, 0.f};
        }
      }
    }
  }
#undef P8P_SET_TILE
#undef P8P_TILE
#undef P8P_READ_A
#undef P8P_READ_B
#undef P8_LDS
#undef P8_LDS2
#undef P8_PHASE
#undef P8_MFMA
#undef P8_READ_A
#undef P8_READ_B
#undef P8_STAGE
#undef P8_BAR
#undef P8_LGKM0
#undef P8_VM
}

#include <cstdlib>

// persistent launch: main P=256 x cnt tiles + remainder P=T%256 x 1
static void gemm_nt_p8p_dispatch(const unsigned short* A, const unsigned short* B,
                                 const unsigned short* bias, unsigned short* C,
                                 unsigned short* Cpre, int M, int N, int K,
                                 int act, hipStream_t stream) {
  const int nbn = N >> 8;
  const int T = (M >> 8) * nbn;
  const int cnt = T / 256;
  const int trem = T - cnt * 256;
#define QN_P8P_ONE(A_, S_, P_, CNT_, T0_)                                      \
  hipLaunchKernelGGL((gemm_nt_p8p_kernel<A_, S_>), dim3(P_), dim3(512), 0,     \
                     stream, A, B, bias, C, Cpre, M, N, K, nbn, CNT_, T0_)
#define QN_P8P_CASE(A_, S_)                                                    \
  do {                                                                         \
    if (cnt > 0) QN_P8P_ONE(A_, S_, 256, cnt, 0);                              \
    if (trem > 0) QN_P8P_ONE(A_, S_, trem, 1, cnt * 256);                      \
  } while (0)
  if (act == QN_ACT_GELU) {
    if (Cpre) QN_P8P_CASE(QN_ACT_GELU, true); else QN_P8P_CASE(QN_ACT_GELU, false);
  } else if (act == QN_ACT_RELU) {
    if (Cpre) QN_P8P_CASE(QN_ACT_RELU, true); else QN_P8P_CASE(QN_ACT_RELU, false);
  } else {
    QN_P8P_CASE(QN_ACT_NONE, false);
  }
#undef QN_P8P_CASE
#undef QN_P8P_ONE
}

void gemm_nt_launch(const unsigned short* A, const unsigned short* B,
                    const unsigned short* bias, unsigned short* C,
                    unsigned short* Cpre, int M, int N, int K, int act,
                    hipStream_t stream, int mode) {
  static int use_glds = -1;
  if (use_glds < 0) {
    const char* e = getenv("QN_GEMM_GLDS");
    use_glds = (e && e[0] == '0') ? 0 : 1;  // default: glds fast path
  }
  static int big_tile = -1;
  if (big_tile < 0) {
    const char* e = getenv("QN_GEMM_BIG");
    big_tile = (e && e[0] == '1') ? 1 : 0;  // 128x128 default (A/B: big tile is a wash at K=768)
  }
  static int use_8p = -1;
  if (use_8p < 0) {
    const char* e = getenv("QN_GEMM_8P");
    use_8p = (e && e[0] == '0') ? 0 : 1;  // default: 8-phase for eligible shapes
  }
  // 8-phase wins only where its 1-block/CU deep pipeline can fill the chip
  // and amortize the 2-tile prologue: ≥256 workgroups and deep K (measured:
  // +22% vs the 128² kernel at 8192³, +5% at 4096³, −20..−40% on the skinny
  // K=768 / N=768 GPT-2 shapes — see profiles/README.md r04).
  const bool p8_ok = (M % 256 == 0) && (N % 256 == 0) && (K % 128 == 0);
  if (mode == 5 && p8_ok) {
    gemm_nt_p8p_dispatch(A, B, bias, C, Cpre, M, N, K, act, stream);
    return;
  }
  const bool p8_shape = p8_ok &&
                        ((long long)(M >> 8) * (N >> 8) >= 256) && (K >= 2048);
  if ((mode == 4 && p8_ok) || (mode == 0 && use_8p && p8_shape)) {
    dim3 grid((M >> 8) * (N >> 8)), blk(512);
#define QN_P8_LAUNCH(A_, S_)                                                   \
  hipLaunchKernelGGL((gemm_nt_8p_kernel<A_, S_>), grid, blk, 0, stream, A, B,  \
                     bias, C, Cpre, M, N, K)
    if (act == QN_ACT_GELU) {
      if (Cpre) QN_P8_LAUNCH(QN_ACT_GELU, true); else QN_P8_LAUNCH(QN_ACT_GELU, false);
    } else if (act == QN_ACT_RELU) {
      if (Cpre) QN_P8_LAUNCH(QN_ACT_RELU, true); else QN_P8_LAUNCH(QN_ACT_RELU, false);
    } else {
      QN_P8_LAUNCH(QN_ACT_NONE, false);
    }
#undef QN_P8_LAUNCH
    return;
  }
  // tile pick: mode 1/2/3 force 128², 256×128, 128×256; mode 0 keeps the
  // measured default (128² unless QN_GEMM_BIG)
  int tile = 1;
  if (mode == 2 && (M % 256 == 0)) tile = 2;
  else if (mode == 3 && (N % 256 == 0)) tile = 3;
  else if (mode == 0 && big_tile && (M % 256 == 0) &&
           ((long long)(M / 256) * ((N + 127) / 128) >= 256)) tile = 2;
  static int occ128 = -1;
  if (occ128 < 0) {
    const char* e = getenv("QN_GEMM_OCC");
    occ128 = e ? atoi(e) : 0;  // 0 = compiler default (2 blocks/CU at 128²)
  }
#define QN_GEMM_LAUNCH(A_, S_, G_, TBM_, TBN_, WR_, WC_)                       \
  do {                                                                         \
    dim3 g_(((M + TBM_ - 1) / TBM_) * ((N + TBN_ - 1) / TBN_));                \
    dim3 b_(WR_ * WC_ * 64);                                                   \
    if (TBM_ == 128 && TBN_ == 128 && occ128 == 3)                             \
      hipLaunchKernelGGL((gemm_nt_kernel<A_, S_, G_, TBM_, TBN_, WR_, WC_, 3>),\
                         g_, b_, 0, stream, A, B, bias, C, Cpre, M, N, K);     \
    else                                                                       \
      hipLaunchKernelGGL((gemm_nt_kernel<A_, S_, G_, TBM_, TBN_, WR_, WC_>),   \
                         g_, b_, 0, stream, A, B, bias, C, Cpre, M, N, K);     \
  } while (0)
#define QN_GEMM_CASE(A_, S_)                                                   \
  do {                                                                         \
    if (tile == 2) {                                                           \
      if (use_glds) QN_GEMM_LAUNCH(A_, S_, true, 256, 128, 4, 2);              \
      else QN_GEMM_LAUNCH(A_, S_, false, 256, 128, 4, 2);                      \
    } else if (tile == 3) {                                                    \
      if (use_glds) QN_GEMM_LAUNCH(A_, S_, true, 128, 256, 2, 4);              \
      else QN_GEMM_LAUNCH(A_, S_, false, 128, 256, 2, 4);                      \
    } else {                                                                   \
      if (use_glds) QN_GEMM_LAUNCH(A_, S_, true, 128, 128, 2, 2);              \
      else QN_GEMM_LAUNCH(A_, S_, false, 128, 128, 2, 2);                      \
    }                                                                          \
  } while (0)
  if (act == QN_ACT_GELU) {
    if (Cpre) QN_GEMM_CASE(QN_ACT_GELU, true); else QN_GEMM_CASE(QN_ACT_GELU, false);
  } else if (act == QN_ACT_RELU) {
    if (Cpre) QN_GEMM_CASE(QN_ACT_RELU, true); else QN_GEMM_CASE(QN_ACT_RELU, false);
  } else {
    QN_GEMM_CASE(QN_ACT_NONE, false);
  }
#undef QN_GEMM_CASE
#undef QN_GEMM_LAUNCH
}
