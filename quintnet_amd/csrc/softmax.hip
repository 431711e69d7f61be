// Fused scale + causal-mask + softmax fwd/bwd for gfx950.
// The softmax half of the attention pipeline (GEMMs ride hipBLASLt;
// SURVEY.md §2.4 "softmax folded into fused attention").
// Rows = B*H*T; row length S (keys). Causal: row r attends to
// cols <= (r % T) + (S - T).
//
// S <= 1024 caches the row in registers (16 f32/lane) — one global read
// instead of three; longer rows fall back to the streaming path.
#include "common.h"

template <typename T, int WPB>
__global__ void softmax_fwd_kernel(
    const T* __restrict__ scores, T* __restrict__ out,
    long long rows, int T_q, int S, float scale, int causal) {
  const int lane = threadIdx.x & (QN_WAVE - 1);
  const int wave = threadIdx.x / QN_WAVE;
  const long long row = (long long)blockIdx.x * WPB + wave;
  if (row >= rows) return;
  const T* sr = scores + row * S;
  T* orow = out + row * S;
  const int t = (int)(row % T_q);
  const int limit = causal ? (t + 1 + (S - T_q)) : S;  // #valid cols

  if (S <= QN_WAVE * 16) {
    float cache[16];
    int n = 0;
    float m = -INFINITY;
    for (int i = lane; i < limit; i += QN_WAVE, ++n) {
      cache[n] = ld_as_f32(sr + i) * scale;
      m = fmaxf(m, cache[n]);
    }
    m = wave_reduce_max(m);
    float sum = 0.f;
    for (int j = 0; j < n; ++j) {
      cache[j] = __expf(cache[j] - m);
      sum += cache[j];
    }
    sum = wave_reduce_sum(sum);
    const float inv = 1.0f / sum;
    n = 0;
    for (int i = lane; i < limit; i += QN_WAVE, ++n)
      st_from_f32(orow + i, cache[n] * inv);
    // zero the masked tail
    for (int i = limit + lane; i < S; i += QN_WAVE)
      st_from_f32(orow + i, 0.f);
  } else {
    float m = -INFINITY;
    for (int i = lane; i < limit; i += QN_WAVE)
      m = fmaxf(m, ld_as_f32(sr + i) * scale);
    m = wave_reduce_max(m);
    float sum = 0.f;
    for (int i = lane; i < limit; i += QN_WAVE)
      sum += __expf(ld_as_f32(sr + i) * scale - m);
    sum = wave_reduce_sum(sum);
    const float inv = 1.0f / sum;
    for (int i = lane; i < S; i += QN_WAVE) {
      float p = (i < limit) ? __expf(ld_as_f32(sr + i) * scale - m) * inv : 0.f;
      st_from_f32(orow + i, p);
    }
  }
}

// dS = scale * P ⊙ (dP - rowsum(dP ⊙ P))
template <typename T, int WPB>
__global__ void softmax_bwd_kernel(
    const T* __restrict__ p, const T* __restrict__ dp, T* __restrict__ ds,
    long long rows, int S, float scale) {
  const int lane = threadIdx.x & (QN_WAVE - 1);
  const int wave = threadIdx.x / QN_WAVE;
  const long long row = (long long)blockIdx.x * WPB + wave;
  if (row >= rows) return;
  const T* pr = p + row * S;
  const T* dpr = dp + row * S;
  T* dsr = ds + row * S;

  if (S <= QN_WAVE * 16) {
    float cp[16], cdp[16];
    int n = 0;
    float dot = 0.f;
    for (int i = lane; i < S; i += QN_WAVE, ++n) {
      cp[n] = ld_as_f32(pr + i);
      cdp[n] = ld_as_f32(dpr + i);
      dot += cp[n] * cdp[n];
    }
    dot = wave_reduce_sum(dot);
    n = 0;
    for (int i = lane; i < S; i += QN_WAVE, ++n)
      st_from_f32(dsr + i, scale * cp[n] * (cdp[n] - dot));
  } else {
    float dot = 0.f;
    for (int i = lane; i < S; i += QN_WAVE)
      dot += ld_as_f32(pr + i) * ld_as_f32(dpr + i);
    dot = wave_reduce_sum(dot);
    for (int i = lane; i < S; i += QN_WAVE) {
      float pv = ld_as_f32(pr + i);
      float dpv = ld_as_f32(dpr + i);
      st_from_f32(dsr + i, scale * pv * (dpv - dot));
    }
  }
}

template <typename T>
void softmax_fwd_launch(const T* scores, T* out, long long rows, int T_q, int S,
                        float scale, int causal, hipStream_t stream) {
  constexpr int WPB = 4;
  dim3 grid((unsigned)((rows + WPB - 1) / WPB));
  hipLaunchKernelGGL((softmax_fwd_kernel<T, WPB>), grid, dim3(WPB * QN_WAVE), 0,
                     stream, scores, out, rows, T_q, S, scale, causal);
}

template <typename T>
void softmax_bwd_launch(const T* p, const T* dp, T* ds, long long rows, int S,
                        float scale, hipStream_t stream) {
  constexpr int WPB = 4;
  dim3 grid((unsigned)((rows + WPB - 1) / WPB));
  hipLaunchKernelGGL((softmax_bwd_kernel<T, WPB>), grid, dim3(WPB * QN_WAVE), 0,
                     stream, p, dp, ds, rows, S, scale);
}

template void softmax_fwd_launch<float>(const float*, float*, long long, int, int, float, int, hipStream_t);
template void softmax_fwd_launch<unsigned short>(const unsigned short*, unsigned short*, long long, int, int, float, int, hipStream_t);
template void softmax_bwd_launch<float>(const float*, const float*, float*, long long, int, float, hipStream_t);
template void softmax_bwd_launch<unsigned short>(const unsigned short*, const unsigned short*, unsigned short*, long long, int, float, hipStream_t);
