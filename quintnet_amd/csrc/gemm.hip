// Hand-written CDNA4 (gfx950) bf16 MFMA GEMM — the TP-linear forward path.
//
//   C[M,N] = act(A[M,K] · B[N,K]^T + bias)        (nn.Linear layout, NT)
//
// Design (guide: cdna_hip_programming.md §5):
//  * 128×128 block tile, BK=64; 4 waves (2×2), each computing a 64×64
//    sub-tile as 4×4 fragments of v_mfma_f32_16x16x32_bf16 (fp32 accum).
//  * A/B tiles staged through LDS in [row][k] layout with the T2 XOR
//    swizzle on 8-element (16 B) groups — ds_read_b128 fragment reads are
//    bank-conflict-free; both MFMA operands read identically since the
//    NT layout makes B's fragment a row of Bs.
//  * T14 async-stage split: next tile's global loads issue right after
//    the barrier, before the MFMA cluster, so HBM latency hides under
//    compute; bounds-checked staging handles arbitrary M/N (K%8==0).
//  * Bias add + GELU/ReLU fused into the epilogue (optionally emitting
//    the pre-activation for backward).
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

// staging: 4 passes × 256 threads × 8 bf16 cover a 128×64 tile
// pass p, thread t -> lin = p*256+t; row = lin>>3, group g = lin&7
template <int PASSES>
__device__ __forceinline__ void load_tile_regs(
    const unsigned short* __restrict__ src, long long ld, int row0, int rows,
    int k0, int K, s16x8 (&regs)[PASSES]) {
#pragma unroll
  for (int p = 0; p < PASSES; ++p) {
    int lin = p * 256 + threadIdx.x;
    int row = lin >> 3;
    int g = lin & 7;
    int k = k0 + g * 8;
    bool ok = (row0 + row < rows) && (k < K);
    regs[p] = ok ? *reinterpret_cast<const s16x8*>(src + (long long)(row0 + row) * ld + k)
                 : zero8();
  }
}

template <int PASSES>
__device__ __forceinline__ void write_tile_lds(unsigned short* lds, const s16x8 (&regs)[PASSES]) {
#pragma unroll
  for (int p = 0; p < PASSES; ++p) {
    int lin = p * 256 + threadIdx.x;
    int row = lin >> 3;
    int g = lin & 7;
    int gs = g ^ (row & 7);  // T2 XOR swizzle on 16B groups
    *reinterpret_cast<s16x8*>(lds + row * BK + gs * 8) = regs[p];
  }
}

template <int ACT, bool SAVE_PRE>
__global__ __launch_bounds__(256) void gemm_nt_kernel(
    const unsigned short* __restrict__ A,  // [M,K]
    const unsigned short* __restrict__ B,  // [N,K]
    const unsigned short* __restrict__ bias,  // [N] or nullptr
    unsigned short* __restrict__ C,        // [M,N]
    unsigned short* __restrict__ Cpre,     // [M,N] pre-activation (SAVE_PRE)
    int M, int N, int K) {
  __shared__ unsigned short As[BM * BK];
  __shared__ unsigned short Bs[BN * BK];

  // XCD-aware bijective block swizzle (T1)
  const int nbm = (M + BM - 1) / BM;
  const int nbn = (N + BN - 1) / BN;
  const int nwg = nbm * nbn;
  int bid = blockIdx.x;
  if (nwg >= 16) {
    const int nx = 8;
    int q = nwg / nx, r = nwg % nx;
    int xcd = bid % nx, idx = bid / nx;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int bm = bid / nbn;
  const int bn = bid % nbn;
  const int m0 = bm * BM;
  const int n0 = bn * BN;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wr = wave >> 1;   // 0..1
  const int wc = wave & 1;    // 0..1

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  s16x8 ra[4], rb[4];
  load_tile_regs<4>(A, K, m0, M, 0, K, ra);
  load_tile_regs<4>(B, K, n0, N, 0, K, rb);

  const int frow = lane & 15;       // fragment row/col within 16
  const int kq = lane >> 4;         // k quarter (0..3) of the 32-K step

  for (int k0 = 0; k0 < K; k0 += BK) {
    __syncthreads();  // previous tile's LDS reads done
    write_tile_lds<4>(As, ra);
    write_tile_lds<4>(Bs, rb);
    __syncthreads();  // tile visible

    // T14: issue next tile's loads before the MFMA cluster
    if (k0 + BK < K) {
      load_tile_regs<4>(A, K, m0, M, k0 + BK, K, ra);
      load_tile_regs<4>(B, K, n0, N, k0 + BK, K, rb);
    }

#pragma unroll
    for (int ks = 0; ks < BK / 32; ++ks) {
      bf16x8 af[4], bf[4];
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        int arow = wr * 64 + f * 16 + frow;
        int ag = (ks * 4 + kq) ^ (arow & 7);
        af[f] = *reinterpret_cast<const bf16x8*>(&As[arow * BK + ag * 8]);
        int brow = wc * 64 + f * 16 + frow;
        int bg = (ks * 4 + kq) ^ (brow & 7);
        bf[f] = *reinterpret_cast<const bf16x8*>(&Bs[brow * BK + bg * 8]);
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[i], bf[j], acc[i][j], 0, 0, 0);
    }
  }

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

void gemm_nt_launch(const unsigned short* A, const unsigned short* B,
                    const unsigned short* bias, unsigned short* C,
                    unsigned short* Cpre, int M, int N, int K, int act,
                    hipStream_t stream) {
  const int nbm = (M + BM - 1) / BM;
  const int nbn = (N + BN - 1) / BN;
  dim3 grid(nbm * nbn);
  dim3 block(256);
#define QN_GEMM_CASE(A_, S_)                                                   \
  hipLaunchKernelGGL((gemm_nt_kernel<A_, S_>), grid, block, 0, stream, A, B,   \
                     bias, C, Cpre, M, N, K)
  if (act == QN_ACT_GELU) {
    if (Cpre) QN_GEMM_CASE(QN_ACT_GELU, true); else QN_GEMM_CASE(QN_ACT_GELU, false);
  } else if (act == QN_ACT_RELU) {
    if (Cpre) QN_GEMM_CASE(QN_ACT_RELU, true); else QN_GEMM_CASE(QN_ACT_RELU, false);
  } else {
    QN_GEMM_CASE(QN_ACT_NONE, false);
  }
#undef QN_GEMM_CASE
}
