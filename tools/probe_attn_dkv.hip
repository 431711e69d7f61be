// Ablation probe for attn_bwd_dkv (NOT linked into the library).
// Variants cut one phase each; asm keep-alives prevent DCE (guide rule 17).
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include "../quintnet_amd/csrc/common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
#define LOG2E 1.4426950408889634f
#define TPAD 40

__device__ __forceinline__ unsigned cvt_pk_bf16(float lo, float hi) {
  unsigned r;
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}
__device__ __forceinline__ bf16x8 relayout8(const float* p) {
  unsigned a0 = cvt_pk_bf16(p[0], p[1]);
  unsigned a1 = cvt_pk_bf16(p[2], p[3]);
  unsigned a2 = cvt_pk_bf16(p[4], p[5]);
  unsigned a3 = cvt_pk_bf16(p[6], p[7]);
  { auto r = __builtin_amdgcn_permlane32_swap(a0, a2, false, false); a0 = r[0]; a2 = r[1]; }
  { auto r = __builtin_amdgcn_permlane32_swap(a1, a3, false, false); a1 = r[0]; a3 = r[1]; }
  union { unsigned u[4]; bf16x8 v; } out;
  out.u[0]=a0; out.u[1]=a1; out.u[2]=a2; out.u[3]=a3;
  return out.v;
}
__device__ __forceinline__ bf16x8 frag_ld(const unsigned short* base, long long rs,
                                          int row0, int d0, int lane) {
  return *reinterpret_cast<const bf16x8*>(base + (long long)(row0 + (lane & 31)) * rs + d0 + ((lane >> 5) << 3));
}
__device__ __forceinline__ int drow(int r, int lane) {
  return (r & 3) + ((r >> 2) << 3) + ((lane >> 5) << 2);
}
__device__ __forceinline__ s16x8 stage_ld(const unsigned short* src, long long rs, int row0) {
  int r = threadIdx.x & 31, d0 = (threadIdx.x >> 5) << 3;
  return *reinterpret_cast<const s16x8*>(src + (long long)(row0 + r) * rs + d0);
}
__device__ __forceinline__ void stage_wr(unsigned short* t, s16x8 v) {
  int r = threadIdx.x & 31, d0 = (threadIdx.x >> 5) << 3;
#pragma unroll
  for (int j = 0; j < 8; ++j) t[(d0 + j) * TPAD + r] = (unsigned short)v[j];
}
#define KEEP8(v) asm volatile("" :: "v"(v[0]), "v"(v[4]))

// CUT: 0 full | 1 no dv/dk mfma+relayout | 2 no softmax VALU | 3 no stage writes | 4 no S/dP mfma
template <int CUT>
__global__ __launch_bounds__(256) void dkv_probe(
    const unsigned short* __restrict__ q, const unsigned short* __restrict__ k,
    const unsigned short* __restrict__ v, const unsigned short* __restrict__ dout,
    const float* __restrict__ lse2, const float* __restrict__ delta,
    unsigned short* __restrict__ dk, unsigned short* __restrict__ dv,
    int T, int H, float scale) {
  __shared__ unsigned short dot_lds[64 * TPAD];
  __shared__ unsigned short qt_lds[64 * TPAD];
  __shared__ float lse_t[32], del_t[32];
  const int bh = blockIdx.y;
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  const int kv0b = blockIdx.x * 128, kw = kv0b + wave * 32;
  const int mykey = kw + (lane & 31);
  const long long off = (long long)bh * T * 64;
  const unsigned short *qp = q + off, *kp = k + off, *vp = v + off, *dop = dout + off;
  unsigned short *dkp = dk + off, *dvp = dv + off;
  bf16x8 kf[4], vf[4];
#pragma unroll
  for (int t = 0; t < 4; ++t) { kf[t] = frag_ld(kp, 64, kw, t*16, lane); vf[t] = frag_ld(vp, 64, kw, t*16, lane); }
  const float s2scale = scale * LOG2E;
  f32x16 dka[2], dva[2], zc;
#pragma unroll
  for (int i = 0; i < 16; ++i) { dka[0][i]=dka[1][i]=dva[0][i]=dva[1][i]=0.f; zc[i]=0.f; }
  bf16x8 qf_n[4], dof_n[4];
  s16x8 dost_n = stage_ld(dop, 64, kv0b), qst_n = stage_ld(qp, 64, kv0b);
  float lse_n = 0.f, del_n = 0.f;
  if (threadIdx.x < 32) { lse_n = lse2[(long long)bh*T + kv0b + threadIdx.x]; del_n = delta[(long long)bh*T + kv0b + threadIdx.x]; }
#pragma unroll
  for (int t = 0; t < 4; ++t) { qf_n[t] = frag_ld(qp, 64, kv0b, t*16, lane); dof_n[t] = frag_ld(dop, 64, kv0b, t*16, lane); }
  for (int qt0 = kv0b; qt0 < T; qt0 += 32) {
    __syncthreads();
    if (CUT != 3) { stage_wr(dot_lds, dost_n); stage_wr(qt_lds, qst_n); }
    else { KEEP8(dost_n); KEEP8(qst_n); }
    if (threadIdx.x < 32) { lse_t[threadIdx.x] = lse_n; del_t[threadIdx.x] = del_n; }
    __syncthreads();
    bf16x8 qf_c[4], dof_c[4];
#pragma unroll
    for (int t = 0; t < 4; ++t) { qf_c[t] = qf_n[t]; dof_c[t] = dof_n[t]; }
    if (qt0 + 32 < T) {
      dost_n = stage_ld(dop, 64, qt0+32); qst_n = stage_ld(qp, 64, qt0+32);
      if (threadIdx.x < 32) { lse_n = lse2[(long long)bh*T + qt0+32+threadIdx.x]; del_n = delta[(long long)bh*T + qt0+32+threadIdx.x]; }
#pragma unroll
      for (int t = 0; t < 4; ++t) { qf_n[t] = frag_ld(qp, 64, qt0+32, t*16, lane); dof_n[t] = frag_ld(dop, 64, qt0+32, t*16, lane); }
    }
    if (qt0 + 31 < kw) continue;
    f32x16 s, dp_;
    if (CUT != 4) {
      s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qf_c[0], kf[0], zc, 0, 0, 0);
      dp_ = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dof_c[0], vf[0], zc, 0, 0, 0);
#pragma unroll
      for (int t = 1; t < 4; ++t) {
        s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qf_c[t], kf[t], s, 0, 0, 0);
        dp_ = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dof_c[t], vf[t], dp_, 0, 0, 0);
      }
    } else {
      s = zc; dp_ = zc;
      KEEP8(qf_c[0]); KEEP8(dof_c[0]);
    }
    const bool diag = (qt0 < kw + 31);
    float pv[16], gv[16];
    if (CUT != 2) {
      if (diag) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int qrow = drow(r, lane);
          float p = (mykey > qt0 + qrow) ? 0.f : __builtin_amdgcn_exp2f(s[r]*s2scale - lse_t[qrow]);
          pv[r] = p; gv[r] = scale * p * (dp_[r] - del_t[qrow]);
        }
      } else {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int qrow = drow(r, lane);
          float p = __builtin_amdgcn_exp2f(s[r]*s2scale - lse_t[qrow]);
          pv[r] = p; gv[r] = scale * p * (dp_[r] - del_t[qrow]);
        }
      }
    } else {
#pragma unroll
      for (int r = 0; r < 16; ++r) { pv[r] = s[r]; gv[r] = dp_[r]; }
    }
    if (CUT != 1) {
      bf16x8 pf0 = relayout8(pv), pf1 = relayout8(pv + 8);
      bf16x8 gf0 = relayout8(gv), gf1 = relayout8(gv + 8);
#pragma unroll
      for (int mt = 0; mt < 2; ++mt) {
        const unsigned short* adot = &dot_lds[(mt*32 + (lane&31))*TPAD + ((lane>>5)<<3)];
        const unsigned short* aqt = &qt_lds[(mt*32 + (lane&31))*TPAD + ((lane>>5)<<3)];
        bf16x8 d0v = *reinterpret_cast<const bf16x8*>(adot);
        bf16x8 d1v = *reinterpret_cast<const bf16x8*>(adot + 16);
        bf16x8 q0f = *reinterpret_cast<const bf16x8*>(aqt);
        bf16x8 q1f = *reinterpret_cast<const bf16x8*>(aqt + 16);
        dva[mt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(d0v, pf0, dva[mt], 0, 0, 0);
        dva[mt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(d1v, pf1, dva[mt], 0, 0, 0);
        dka[mt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(q0f, gf0, dka[mt], 0, 0, 0);
        dka[mt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(q1f, gf1, dka[mt], 0, 0, 0);
      }
    } else {
      asm volatile("" :: "v"(pv[0]), "v"(gv[0]), "v"(pv[8]), "v"(gv[8]));
    }
  }
#pragma unroll
  for (int mt = 0; mt < 2; ++mt)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int d = mt*32 + drow(r, lane);
      dkp[(long long)mykey*64 + d] = f32_to_bf16(dka[mt][r]);
      dvp[(long long)mykey*64 + d] = f32_to_bf16(dva[mt][r]);
    }
}

int main() {
  int B = 8, H = 12, T = 1024;
  long long n = (long long)B*H*T*64;
  unsigned short *q, *k, *v, *dgo, *dk, *dv; float *lse, *del;
  hipMalloc(&q, n*2); hipMalloc(&k, n*2); hipMalloc(&v, n*2); hipMalloc(&dgo, n*2);
  hipMalloc(&dk, n*2); hipMalloc(&dv, n*2);
  hipMalloc(&lse, (long long)B*H*T*4); hipMalloc(&del, (long long)B*H*T*4);
  hipMemset(q, 0x3c, n*2); hipMemset(k, 0x3c, n*2); hipMemset(v, 0x3c, n*2); hipMemset(dgo, 0x3c, n*2);
  hipMemset(lse, 0x3f, (long long)B*H*T*4); hipMemset(del, 0x3f, (long long)B*H*T*4);
  dim3 grid(T/128, B*H), blk(256);
  hipEvent_t e0, e1; hipEventCreate(&e0); hipEventCreate(&e1);
#define RUN(CUT, name)                                                        \
  {                                                                           \
    for (int i = 0; i < 3; ++i)                                               \
      hipLaunchKernelGGL((dkv_probe<CUT>), grid, blk, 0, 0, q, k, v, dgo, lse, del, dk, dv, T, H, 0.125f); \
    hipEventRecord(e0);                                                       \
    for (int i = 0; i < 20; ++i)                                              \
      hipLaunchKernelGGL((dkv_probe<CUT>), grid, blk, 0, 0, q, k, v, dgo, lse, del, dk, dv, T, H, 0.125f); \
    hipEventRecord(e1); hipEventSynchronize(e1);                              \
    float ms; hipEventElapsedTime(&ms, e0, e1);                               \
    printf("%-18s %8.1f us\n", name, ms * 1000 / 20);                         \
  }
  RUN(0, "full")
  RUN(1, "no-dvdk-mfma")
  RUN(2, "no-softmax-valu")
  RUN(3, "no-stage-writes")
  RUN(4, "no-sdp-mfma")
  return 0;
}
