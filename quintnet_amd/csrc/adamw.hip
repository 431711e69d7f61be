// Fused AdamW over a flat shard (ZeRO-1 local update) + multi-tensor
// sum-of-squares for grad clipping. fp32 master/m/v; params bf16 or f32.
// Replaces the eager optim.AdamW loops (reference GPT2_Trainer.py:100).
#include "common.h"

template <typename TP, typename TG>
__global__ void adamw_kernel(
    TP* __restrict__ param, float* __restrict__ master, const TG* __restrict__ grad,
    float* __restrict__ m, float* __restrict__ v,
    const long long* __restrict__ step_dev,
    long long n, float lr, float beta1, float beta2, float eps, float wd,
    float bc1, float bc2) {
  if (step_dev) {  // hipGraph-replay path: bias correction from device step
    float st = (float)*step_dev;
    bc1 = 1.f - __powf(beta1, st);
    bc2 = 1.f - __powf(beta2, st);
  }
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < n; i += stride) {
    float g = ld_as_f32(grad + i);
    float mi = m[i] = beta1 * m[i] + (1.f - beta1) * g;
    float vi = v[i] = beta2 * v[i] + (1.f - beta2) * g * g;
    float denom = sqrtf(vi / bc2) + eps;
    float p = master[i];
    p *= (1.f - lr * wd);                 // decoupled weight decay
    p -= lr * (mi / bc1) / denom;
    master[i] = p;
    st_from_f32(param + i, p);
  }
}

template <typename T>
__global__ void sumsq_kernel(const T* __restrict__ x, long long n, float* __restrict__ out) {
  __shared__ float scratch[8];
  float acc = 0.f;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < n; i += stride) {
    float v = ld_as_f32(x + i);
    acc += v * v;
  }
  acc = block_reduce_sum<4>(acc, scratch);
  if (threadIdx.x == 0) atomicAdd(out, acc);
}

template <typename TP, typename TG>
void adamw_launch(TP* param, float* master, const TG* grad, float* m, float* v,
                  const long long* step_dev, long long n, float lr, float beta1,
                  float beta2, float eps, float wd, int step, hipStream_t stream) {
  float bc1 = 1.f - powf(beta1, (float)step);
  float bc2 = 1.f - powf(beta2, (float)step);
  long long blocks = min((n + 255) / 256, (long long)2048);
  hipLaunchKernelGGL((adamw_kernel<TP, TG>), dim3((unsigned)blocks), dim3(256), 0,
                     stream, param, master, grad, m, v, step_dev, n, lr, beta1,
                     beta2, eps, wd, bc1, bc2);
}

template <typename T>
void sumsq_launch(const T* x, long long n, float* out, hipStream_t stream) {
  long long blocks = min((n + 255) / 256, (long long)2048);
  hipLaunchKernelGGL((sumsq_kernel<T>), dim3((unsigned)blocks), dim3(256), 0, stream, x, n, out);
}

template void adamw_launch<float, float>(float*, float*, const float*, float*, float*, const long long*, long long, float, float, float, float, float, int, hipStream_t);
template void adamw_launch<unsigned short, unsigned short>(unsigned short*, float*, const unsigned short*, float*, float*, const long long*, long long, float, float, float, float, float, int, hipStream_t);
template void adamw_launch<unsigned short, float>(unsigned short*, float*, const float*, float*, float*, const long long*, long long, float, float, float, float, float, int, hipStream_t);
template void adamw_launch<float, unsigned short>(float*, float*, const unsigned short*, float*, float*, const long long*, long long, float, float, float, float, float, int, hipStream_t);
template void sumsq_launch<float>(const float*, long long, float*, hipStream_t);
template void sumsq_launch<unsigned short>(const unsigned short*, long long, float*, hipStream_t);
