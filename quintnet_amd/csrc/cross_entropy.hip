// Fused cross-entropy (log-softmax + NLL) over a large vocab (50257).
// One 4-wave block per row, grid-stride over rows; fp32 accumulation;
// fwd emits per-row lse + atomic loss sum, bwd writes
// (softmax - onehot)/n_valid in one pass. Replaces nn.CrossEntropyLoss
// (reference trainer.py:90, GPT2_Trainer.py:109).
#include "common.h"

template <typename T>
__global__ void ce_fwd_kernel(
    const T* __restrict__ logits, const long long* __restrict__ target,
    float* __restrict__ lse_out, float* __restrict__ loss_sum,
    long long rows, int V, long long ignore_index) {
  __shared__ float scratch[8];
  for (long long row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* lr = logits + row * V;
    const long long tgt = target[row];
    float m = -INFINITY;
    for (int i = threadIdx.x; i < V; i += blockDim.x)
      m = fmaxf(m, ld_as_f32(lr + i));
    m = block_reduce_max<4>(m, scratch);
    __syncthreads();
    float sum = 0.f;
    for (int i = threadIdx.x; i < V; i += blockDim.x)
      sum += __expf(ld_as_f32(lr + i) - m);
    sum = block_reduce_sum<4>(sum, scratch);
    const float lse = m + __logf(sum);
    if (threadIdx.x == 0) {
      lse_out[row] = lse;
      if (tgt != ignore_index) {
        float nll = lse - ld_as_f32(lr + (int)tgt);
        atomicAdd(loss_sum, nll);
      }
    }
    __syncthreads();
  }
}

template <typename T>
__global__ void ce_bwd_kernel(
    const T* __restrict__ logits, const long long* __restrict__ target,
    const float* __restrict__ lse, T* __restrict__ dlogits,
    long long rows, int V, long long ignore_index, float inv_n) {
  for (long long row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* lr = logits + row * V;
    T* dr = dlogits + row * V;
    const long long tgt = target[row];
    if (tgt == ignore_index) {
      for (int i = threadIdx.x; i < V; i += blockDim.x) st_from_f32(dr + i, 0.f);
    } else {
      const float l = lse[row];
      for (int i = threadIdx.x; i < V; i += blockDim.x) {
        float p = __expf(ld_as_f32(lr + i) - l);
        if ((long long)i == tgt) p -= 1.0f;
        st_from_f32(dr + i, p * inv_n);
      }
    }
  }
}

template <typename T>
void ce_fwd_launch(const T* logits, const long long* target, float* lse,
                   float* loss_sum, long long rows, int V, long long ignore_index,
                   hipStream_t stream) {
  int grid = (int)min((long long)2048, rows);
  hipLaunchKernelGGL((ce_fwd_kernel<T>), dim3(grid), dim3(256), 0, stream,
                     logits, target, lse, loss_sum, rows, V, ignore_index);
}

template <typename T>
void ce_bwd_launch(const T* logits, const long long* target, const float* lse,
                   T* dlogits, long long rows, int V, long long ignore_index,
                   float inv_n, hipStream_t stream) {
  int grid = (int)min((long long)2048, rows);
  hipLaunchKernelGGL((ce_bwd_kernel<T>), dim3(grid), dim3(256), 0, stream,
                     logits, target, lse, dlogits, rows, V, ignore_index, inv_n);
}

template void ce_fwd_launch<float>(const float*, const long long*, float*, float*, long long, int, long long, hipStream_t);
template void ce_fwd_launch<unsigned short>(const unsigned short*, const long long*, float*, float*, long long, int, long long, hipStream_t);
template void ce_bwd_launch<float>(const float*, const long long*, const float*, float*, long long, int, long long, float, hipStream_t);
template void ce_bwd_launch<unsigned short>(const unsigned short*, const long long*, const float*, unsigned short*, long long, int, long long, float, hipStream_t);
