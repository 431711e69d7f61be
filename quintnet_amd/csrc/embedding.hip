// Fused token+position embedding lookup (gather) and its scatter-add
// backward for gfx950.  out[n] = wte[ids[n]] + wpe[n % T] in one pass —
// replaces the two eager gathers + add of reference
// utils/GPT2/gpt2_embeddings.py:92-95 (SURVEY.md §2.4 "embedding
// lookup / scatter-add").
#include "common.h"

template <typename T>
__global__ void embedding_pair_fwd_kernel(
    const long long* __restrict__ ids, const T* __restrict__ wte,
    const T* __restrict__ wpe, T* __restrict__ out, long long rows, int Tlen,
    int H) {
  const int lane = threadIdx.x & (QN_WAVE - 1);
  const int wave = threadIdx.x / QN_WAVE;
  const long long row = (long long)blockIdx.x * 4 + wave;
  if (row >= rows) return;
  const long long tok = ids[row];
  const int pos = (int)(row % Tlen);
  const T* te = wte + tok * H;
  const T* pe = wpe + (long long)pos * H;
  T* o = out + row * H;
  if constexpr (sizeof(T) == 2) {
    for (int i = lane * 8; i < H; i += QN_WAVE * 8) {
      s16x8 a = *reinterpret_cast<const s16x8*>(te + i);
      s16x8 b = *reinterpret_cast<const s16x8*>(pe + i);
      s16x8 r;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        r[j] = (short)f32_to_bf16(bf16_to_f32((unsigned short)a[j]) +
                                  bf16_to_f32((unsigned short)b[j]));
      *reinterpret_cast<s16x8*>(o + i) = r;
    }
    for (int i = (H & ~7) + lane; i < H; i += QN_WAVE)
      st_from_f32(o + i, ld_as_f32(te + i) + ld_as_f32(pe + i));
  } else {
    for (int i = lane; i < H; i += QN_WAVE)
      st_from_f32(o + i, ld_as_f32(te + i) + ld_as_f32(pe + i));
  }
}

// backward: dwte[ids[n]] += dout[n]; dwpe[n % T] += dout[n]  (fp32 atomics)
template <typename T>
__global__ void embedding_pair_bwd_kernel(
    const long long* __restrict__ ids, const T* __restrict__ dout,
    float* __restrict__ dwte, float* __restrict__ dwpe, long long rows,
    int Tlen, int H) {
  const int lane = threadIdx.x & (QN_WAVE - 1);
  const int wave = threadIdx.x / QN_WAVE;
  const long long row = (long long)blockIdx.x * 4 + wave;
  if (row >= rows) return;
  const long long tok = ids[row];
  const int pos = (int)(row % Tlen);
  const T* g = dout + row * H;
  float* te = dwte + tok * H;
  float* pe = dwpe + (long long)pos * H;
  for (int i = lane; i < H; i += QN_WAVE) {
    float v = ld_as_f32(g + i);
    atomicAdd(te + i, v);
    atomicAdd(pe + i, v);
  }
}

template <typename T>
void embedding_pair_fwd_launch(const long long* ids, const T* wte, const T* wpe,
                               T* out, long long rows, int Tlen, int H,
                               hipStream_t stream) {
  hipLaunchKernelGGL((embedding_pair_fwd_kernel<T>),
                     dim3((unsigned)((rows + 3) / 4)), dim3(256), 0, stream,
                     ids, wte, wpe, out, rows, Tlen, H);
}

template <typename T>
void embedding_pair_bwd_launch(const long long* ids, const T* dout, float* dwte,
                               float* dwpe, long long rows, int Tlen, int H,
                               hipStream_t stream) {
  hipLaunchKernelGGL((embedding_pair_bwd_kernel<T>),
                     dim3((unsigned)((rows + 3) / 4)), dim3(256), 0, stream,
                     ids, dout, dwte, dwpe, rows, Tlen, H);
}

template void embedding_pair_fwd_launch<float>(const long long*, const float*, const float*, float*, long long, int, int, hipStream_t);
template void embedding_pair_fwd_launch<unsigned short>(const long long*, const unsigned short*, const unsigned short*, unsigned short*, long long, int, int, hipStream_t);
template void embedding_pair_bwd_launch<float>(const long long*, const float*, float*, float*, long long, int, int, hipStream_t);
template void embedding_pair_bwd_launch<unsigned short>(const long long*, const unsigned short*, float*, float*, long long, int, int, hipStream_t);
