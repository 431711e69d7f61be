// Fused cross-entropy (log-softmax + NLL) over a large vocab (50257).
// One 4-wave block per row, grid-stride over rows; fp32 accumulation.
// fwd is a single ONLINE pass (running max+sum, one read of the 800 MB
// logits instead of two); bwd writes grad_scale*(softmax - onehot)/n in
// one pass with the upstream grad scale fused (the separate eager
// multiply was 2.5% of the GPT-2 step — rocprof r01).
#include "common.h"

__device__ __forceinline__ void online_combine(float& m, float& s, float m2, float s2) {
  float mn = fmaxf(m, m2);
  // expf(-inf - -inf) guards: if both -inf, s stays 0
  s = ((m == -INFINITY) ? 0.f : s * __expf(m - mn)) +
      ((m2 == -INFINITY) ? 0.f : s2 * __expf(m2 - mn));
  m = mn;
}

template <typename T>
__global__ void ce_fwd_kernel(
    const T* __restrict__ logits, const long long* __restrict__ target,
    float* __restrict__ lse_out, float* __restrict__ loss_sum,
    long long rows, int V, long long ignore_index) {
  __shared__ float sm[8], ss[8];
  const int lane = threadIdx.x & (QN_WAVE - 1);
  const int wave = threadIdx.x / QN_WAVE;
  const int V8 = V >> 3;
  for (long long row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* lr = logits + row * V;
    // 8-wide loads (G13) + 4 independent online partials for ILP
    float m4[4] = {-INFINITY, -INFINITY, -INFINITY, -INFINITY};
    float s4[4] = {0.f, 0.f, 0.f, 0.f};
    if constexpr (sizeof(T) == 2) {
      for (int gI = threadIdx.x; gI < V8; gI += blockDim.x) {
        s16x8 vv = *reinterpret_cast<const s16x8*>(lr + gI * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float x = bf16_to_f32((unsigned short)vv[j]);
          int a = j & 3;
          if (x > m4[a]) { s4[a] = s4[a] * __expf(m4[a] - x) + 1.f; m4[a] = x; }
          else s4[a] += __expf(x - m4[a]);
        }
      }
    } else {
      for (int gI = threadIdx.x; gI < V8; gI += blockDim.x) {
        f32x4 v0 = *reinterpret_cast<const f32x4*>(lr + gI * 8);
        f32x4 v1 = *reinterpret_cast<const f32x4*>(lr + gI * 8 + 4);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float x = (j < 4) ? v0[j & 3] : v1[j & 3];
          int a = j & 3;
          if (x > m4[a]) { s4[a] = s4[a] * __expf(m4[a] - x) + 1.f; m4[a] = x; }
          else s4[a] += __expf(x - m4[a]);
        }
      }
    }
    // vocab tail
    for (int i = V8 * 8 + threadIdx.x; i < V; i += blockDim.x) {
      float x = ld_as_f32(lr + i);
      if (x > m4[0]) { s4[0] = s4[0] * __expf(m4[0] - x) + 1.f; m4[0] = x; }
      else s4[0] += __expf(x - m4[0]);
    }
    float m = m4[0], s = s4[0];
#pragma unroll
    for (int a = 1; a < 4; ++a) online_combine(m, s, m4[a], s4[a]);
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      online_combine(m, s, __shfl_xor(m, off, QN_WAVE), __shfl_xor(s, off, QN_WAVE));
    if (lane == 0) { sm[wave] = m; ss[wave] = s; }
    __syncthreads();
    if (threadIdx.x == 0) {
      float M = sm[0], S = ss[0];
      const int nwaves = blockDim.x / QN_WAVE;
      for (int i = 1; i < nwaves; ++i) online_combine(M, S, sm[i], ss[i]);
      float lse = M + __logf(S);
      lse_out[row] = lse;
      const long long tgt = target[row];
      if (tgt != ignore_index)
        atomicAdd(loss_sum, lse - ld_as_f32(lr + (int)tgt));
    }
    __syncthreads();
  }
}

template <typename T>
__global__ void ce_bwd_kernel(
    const T* __restrict__ logits, const long long* __restrict__ target,
    const float* __restrict__ lse, const float* __restrict__ grad_scale,
    const long long* __restrict__ n_valid, T* __restrict__ dlogits,
    long long rows, int V, long long ignore_index, float inv_n) {
  // n_valid read on-device (no host sync; graph-capture safe)
  const float gs = (grad_scale ? *grad_scale : 1.f) *
                   (n_valid ? 1.f / (float)max(*n_valid, 1ll) : inv_n);
  for (long long row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* lr = logits + row * V;
    T* dr = dlogits + row * V;
    const long long tgt = target[row];
    const int V8 = V >> 3;
    if (tgt == ignore_index) {
      if constexpr (sizeof(T) == 2) {
        s16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
        for (int gI = threadIdx.x; gI < V8; gI += blockDim.x)
          *reinterpret_cast<s16x8*>(dr + gI * 8) = z;
      } else {
        for (int gI = threadIdx.x; gI < V8 * 2; gI += blockDim.x) {
          f32x4 z = {0.f, 0.f, 0.f, 0.f};
          *reinterpret_cast<f32x4*>(dr + gI * 4) = z;
        }
      }
      for (int i = V8 * 8 + threadIdx.x; i < V; i += blockDim.x)
        st_from_f32(dr + i, 0.f);
    } else {
      const float l = lse[row];
      if constexpr (sizeof(T) == 2) {
        for (int gI = threadIdx.x; gI < V8; gI += blockDim.x) {
          s16x8 vv = *reinterpret_cast<const s16x8*>(lr + gI * 8);
          s16x8 ov;
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            float p = __expf(bf16_to_f32((unsigned short)vv[j]) - l);
            if ((long long)(gI * 8 + j) == tgt) p -= 1.0f;
            ov[j] = (short)f32_to_bf16(p * gs);
          }
          *reinterpret_cast<s16x8*>(dr + gI * 8) = ov;
        }
      } else {
        for (int gI = threadIdx.x; gI < V8; gI += blockDim.x) {
#pragma unroll
          for (int half = 0; half < 2; ++half) {
            f32x4 vv = *reinterpret_cast<const f32x4*>(lr + gI * 8 + half * 4);
            f32x4 ov;
#pragma unroll
            for (int j = 0; j < 4; ++j) {
              float p = __expf(vv[j] - l);
              if ((long long)(gI * 8 + half * 4 + j) == tgt) p -= 1.0f;
              ov[j] = p * gs;
            }
            *reinterpret_cast<f32x4*>(dr + gI * 8 + half * 4) = ov;
          }
        }
      }
      for (int i = V8 * 8 + threadIdx.x; i < V; i += blockDim.x) {
        float p = __expf(ld_as_f32(lr + i) - l);
        if ((long long)i == tgt) p -= 1.0f;
        st_from_f32(dr + i, p * gs);
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
                   const float* grad_scale, const long long* n_valid,
                   T* dlogits, long long rows, int V, long long ignore_index,
                   float inv_n, hipStream_t stream) {
  int grid = (int)min((long long)2048, rows);
  hipLaunchKernelGGL((ce_bwd_kernel<T>), dim3(grid), dim3(256), 0, stream,
                     logits, target, lse, grad_scale, n_valid, dlogits, rows,
                     V, ignore_index, inv_n);
}

template void ce_fwd_launch<float>(const float*, const long long*, float*, float*, long long, int, long long, hipStream_t);
template void ce_fwd_launch<unsigned short>(const unsigned short*, const long long*, float*, float*, long long, int, long long, hipStream_t);
template void ce_bwd_launch<float>(const float*, const long long*, const float*, const float*, const long long*, float*, long long, int, long long, float, hipStream_t);
template void ce_bwd_launch<unsigned short>(const unsigned short*, const long long*, const float*, const float*, const long long*, unsigned short*, long long, int, long long, float, hipStream_t);
