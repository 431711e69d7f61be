// Fused elementwise kernels: activation backward (GELU/ReLU) in one pass.
// Replaces the ~10-kernel eager float chain the Python GELU backward cost
// (rocprof r01: tanh/pow/mul/add ≈8% of the GPT-2 step).
#include "common.h"

__device__ __forceinline__ float dgelu_tanh(float x) {
  const float c = 0.7978845608028654f;
  float x2 = x * x;
  float t = tanhf(c * (x + 0.044715f * x * x2));
  return 0.5f * (1.f + t) + 0.5f * x * (1.f - t * t) * c * (1.f + 3.f * 0.044715f * x2);
}

// ACT: 1 = gelu, 2 = relu.  8-wide bf16 path (G13).
template <typename T, int ACT>
__global__ void act_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ pre,
                               T* __restrict__ dx, long long n) {
  const long long stride = (long long)gridDim.x * blockDim.x;
  if constexpr (sizeof(T) == 2) {
    const long long n8 = n >> 3;
    for (long long gI = blockIdx.x * (long long)blockDim.x + threadIdx.x; gI < n8; gI += stride) {
      float gv[8], pv[8], ov[8];
      ld8_f32(reinterpret_cast<const unsigned short*>(dy) + gI * 8, gv);
      ld8_f32(reinterpret_cast<const unsigned short*>(pre) + gI * 8, pv);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        if constexpr (ACT == 1) ov[j] = gv[j] * dgelu_tanh(pv[j]);
        else ov[j] = (pv[j] > 0.f) ? gv[j] : 0.f;
      }
      st8_f32(reinterpret_cast<unsigned short*>(dx) + gI * 8, ov);
    }
    for (long long i = n8 * 8 + blockIdx.x * (long long)blockDim.x + threadIdx.x; i < n; i += stride) {
      float g = ld_as_f32(dy + i);
      float p = ld_as_f32(pre + i);
      float v;
      if constexpr (ACT == 1) v = g * dgelu_tanh(p);
      else v = (p > 0.f) ? g : 0.f;
      st_from_f32(dx + i, v);
    }
  } else {
    for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < n; i += stride) {
      float g = ld_as_f32(dy + i);
      float p = ld_as_f32(pre + i);
      float v;
      if constexpr (ACT == 1) v = g * dgelu_tanh(p);
      else v = (p > 0.f) ? g : 0.f;
      st_from_f32(dx + i, v);
    }
  }
}

template <typename T>
void act_bwd_launch(const T* dy, const T* pre, T* dx, long long n, int act,
                    hipStream_t stream) {
  long long blocks = min((n + 255) / 256, (long long)2048);
  if (act == 1)
    hipLaunchKernelGGL((act_bwd_kernel<T, 1>), dim3((unsigned)blocks), dim3(256), 0, stream, dy, pre, dx, n);
  else
    hipLaunchKernelGGL((act_bwd_kernel<T, 2>), dim3((unsigned)blocks), dim3(256), 0, stream, dy, pre, dx, n);
}

template void act_bwd_launch<float>(const float*, const float*, float*, long long, int, hipStream_t);
template void act_bwd_launch<unsigned short>(const unsigned short*, const unsigned short*, unsigned short*, long long, int, hipStream_t);

// column sum: out[n] = sum_m x[m][n]  (bias gradient; fp32 atomics per
// column-chunk — replaces torch's generic reduce at ~4x the time)
template <typename T>
__global__ void colsum_kernel(const T* __restrict__ x, float* __restrict__ out,
                              long long rows, int N) {
  const int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= N) return;
  const int r0 = blockIdx.y;
  const int nchunks = gridDim.y;
  float acc = 0.f;
  for (long long r = r0; r < rows; r += nchunks)
    acc += ld_as_f32(x + r * N + col);
  atomicAdd(out + col, acc);
}

// bf16 fast path: 8 columns per lane (one 16B dwordx4 load per row — guide
// G13: scalar bf16 loads run at ~2.5x the cost), 4-row unroll for MLP ILP.
// bf16 path is atomic-free: kernel 1 writes per-chunk partials (each block.y
// chunk owns a CONTIGUOUS row range so waves stream sequential lines; 8-deep
// row unroll keeps ≥128B of loads in flight per lane), kernel 2 folds the
// [chunks, N] partial matrix and emits bf16 directly.  Chunk-count formula
// targets ≥64k lanes in kernel 1 regardless of N (small-N bias grads were
// previously serialized on output-line atomics).
__global__ void colsum8_part_kernel(const unsigned short* __restrict__ x,
                                    float* __restrict__ part, long long rows,
                                    int N, long long rpc) {
  const int col0 = (blockIdx.x * blockDim.x + threadIdx.x) * 8;
  if (col0 >= N) return;
  long long r = (long long)blockIdx.y * rpc;
  const long long rend = min(r + rpc, rows);
  float acc[8] = {0.f};
  for (; r + 8 <= rend; r += 8) {
    float a[8][8];
#pragma unroll
    for (int i = 0; i < 8; ++i) ld8_f32(x + (r + i) * N + col0, a[i]);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      acc[j] += ((a[0][j] + a[1][j]) + (a[2][j] + a[3][j])) +
                ((a[4][j] + a[5][j]) + (a[6][j] + a[7][j]));
  }
  for (; r < rend; ++r) {
    float a[8];
    ld8_f32(x + r * N + col0, a);
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] += a[j];
  }
  float* p = part + (long long)blockIdx.y * N + col0;
#pragma unroll
  for (int j = 0; j < 8; ++j) p[j] = acc[j];
}

// one block per column-group-of-8; threads stride the chunk axis, then a
// block reduction per column folds to the final bf16 value
__global__ void colsum8_fold_kernel(const float* __restrict__ part,
                                    unsigned short* __restrict__ out, int N,
                                    int chunks) {
  __shared__ float scratch[8];
  const int col0 = blockIdx.x * 8;
  float acc[8] = {0.f};
  for (int c = threadIdx.x; c < chunks; c += blockDim.x) {
    const float* p = part + (long long)c * N + col0;
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] += p[j];
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) acc[j] = block_reduce_sum<4>(acc[j], scratch);
  if (threadIdx.x < 8) out[col0 + threadIdx.x] = f32_to_bf16(acc[threadIdx.x]);
}

long long colsum_bf16_chunks(long long rows, int N) {
  int colthreads = N / 8;
  long long chunks = (65536 + colthreads - 1) / colthreads;
  chunks = min(max(chunks, (long long)1), min(rows, (long long)2048));
  long long rpc = (rows + chunks - 1) / chunks;
  return (rows + rpc - 1) / rpc;
}

// returns true if the fast path ran; caller provides part = [chunks, N] fp32
bool colsum_bf16_launch(const unsigned short* x, float* part,
                        unsigned short* out, long long rows, int N,
                        hipStream_t stream) {
  if ((N & 7) != 0 || ((uintptr_t)x & 15u) != 0) return false;
  long long chunks = colsum_bf16_chunks(rows, N);
  long long rpc = (rows + chunks - 1) / chunks;
  int colblocks = (N / 8 + 255) / 256;
  hipLaunchKernelGGL(colsum8_part_kernel, dim3(colblocks, (unsigned)chunks),
                     dim3(256), 0, stream, x, part, rows, N, rpc);
  hipLaunchKernelGGL(colsum8_fold_kernel, dim3(N / 8), dim3(256), 0, stream,
                     part, out, N, (int)chunks);
  return true;
}

template <typename T>
void colsum_launch(const T* x, float* out, long long rows, int N, hipStream_t stream) {
  int colblocks = (N + 255) / 256;
  int chunks = (int)min(max(rows / 64, (long long)1), (long long)256);
  hipLaunchKernelGGL((colsum_kernel<T>), dim3(colblocks, chunks), dim3(256), 0,
                     stream, x, out, rows, N);
}

template void colsum_launch<unsigned short>(const unsigned short*, float*, long long, int, hipStream_t);

template void colsum_launch<float>(const float*, float*, long long, int, hipStream_t);

// ---------------------------------------------------------------------------
// fused dropout (counter-based PRNG: mask recomputable, stored as u8 for
// the backward pass).  Replaces eager nn.Dropout (reference
// gpt2_attention.py:108, gpt2_mlp.py:125, gpt2_embeddings.py:59).
__device__ __forceinline__ unsigned wang_hash(unsigned s) {
  s = (s ^ 61u) ^ (s >> 16);
  s *= 9u;
  s = s ^ (s >> 4);
  s *= 0x27d4eb2du;
  s = s ^ (s >> 15);
  return s;
}

template <typename T>
__global__ void dropout_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                   unsigned char* __restrict__ mask,
                                   long long n, float p, float scale,
                                   unsigned seed) {
  const long long stride = (long long)gridDim.x * blockDim.x;
  const unsigned thresh = (unsigned)(p * 4294967296.0f);
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < n; i += stride) {
    unsigned r = wang_hash(seed ^ (unsigned)(i & 0xffffffffu)) ^ wang_hash((unsigned)(i >> 32) + seed * 2654435761u);
    unsigned char keep = r >= thresh;
    mask[i] = keep;
    st_from_f32(y + i, keep ? ld_as_f32(x + i) * scale : 0.f);
  }
}

template <typename T>
__global__ void dropout_bwd_kernel(const T* __restrict__ dy, T* __restrict__ dx,
                                   const unsigned char* __restrict__ mask,
                                   long long n, float scale) {
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < n; i += stride)
    st_from_f32(dx + i, mask[i] ? ld_as_f32(dy + i) * scale : 0.f);
}

template <typename T>
void dropout_fwd_launch(const T* x, T* y, unsigned char* mask, long long n,
                        float p, unsigned seed, hipStream_t stream) {
  long long blocks = min((n + 255) / 256, (long long)2048);
  float scale = 1.0f / (1.0f - p);
  hipLaunchKernelGGL((dropout_fwd_kernel<T>), dim3((unsigned)blocks), dim3(256),
                     0, stream, x, y, mask, n, p, scale, seed);
}

template <typename T>
void dropout_bwd_launch(const T* dy, T* dx, const unsigned char* mask,
                        long long n, float p, hipStream_t stream) {
  long long blocks = min((n + 255) / 256, (long long)2048);
  float scale = 1.0f / (1.0f - p);
  hipLaunchKernelGGL((dropout_bwd_kernel<T>), dim3((unsigned)blocks), dim3(256),
                     0, stream, dy, dx, mask, n, scale);
}

template void dropout_fwd_launch<float>(const float*, float*, unsigned char*, long long, float, unsigned, hipStream_t);
template void dropout_fwd_launch<unsigned short>(const unsigned short*, unsigned short*, unsigned char*, long long, float, unsigned, hipStream_t);
template void dropout_bwd_launch<float>(const float*, float*, const unsigned char*, long long, float, hipStream_t);
template void dropout_bwd_launch<unsigned short>(const unsigned short*, unsigned short*, const unsigned char*, long long, float, hipStream_t);
