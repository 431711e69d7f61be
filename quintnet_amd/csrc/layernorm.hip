// Fused LayerNorm fwd/bwd for gfx950 — one wave per row, fp32 accumulate.
// Replaces the implicit eager LayerNorm of the reference.
// Rows are B*T; H ∈ {64, 768, 3072, ...}. bf16 loads vectorized as s16x8
// (guide G13: scalar bf16 ≈2× slower), one pass of sum/sumsq per row.
#include "common.h"

// ---------------------------------------------------------------------------
// forward: y = (x - mean) * rstd * w + b ; saves mean, rstd (f32 per row)
// grid: (rows / WPB) blocks of WPB waves; wave r handles row r.
// ---------------------------------------------------------------------------
template <typename T, int WPB>
__global__ void layernorm_fwd_kernel(
    const T* __restrict__ x, const T* __restrict__ w, const T* __restrict__ b,
    T* __restrict__ y, float* __restrict__ mean_out, float* __restrict__ rstd_out,
    int rows, int H, float eps) {
  const int lane = threadIdx.x & (QN_WAVE - 1);
  const int wave = threadIdx.x / QN_WAVE;
  const long long row = (long long)blockIdx.x * WPB + wave;
  if (row >= rows) return;
  const T* xr = x + row * H;
  T* yr = y + row * H;

  float sum = 0.f, sq = 0.f;
  // strided per-lane pass (vectorize by 8 when H % (64*8) allows)
  if constexpr (sizeof(T) == 2) {
    if ((H & 511) == 0) {  // H multiple of 512: 8-wide vector loads
      for (int i = lane * 8; i < H; i += QN_WAVE * 8) {
        s16x8 v = *reinterpret_cast<const s16x8*>(xr + i);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = bf16_to_f32((unsigned short)v[j]);
          sum += f; sq += f * f;
        }
      }
    } else {
      for (int i = lane; i < H; i += QN_WAVE) {
        float f = ld_as_f32(reinterpret_cast<const unsigned short*>(xr) + i);
        sum += f; sq += f * f;
      }
    }
  } else {
    for (int i = lane; i < H; i += QN_WAVE) {
      float f = ld_as_f32(xr + i);
      sum += f; sq += f * f;
    }
  }
  sum = wave_reduce_sum(sum);
  sq = wave_reduce_sum(sq);
  const float mean = sum / H;
  const float var = fmaxf(sq / H - mean * mean, 0.f);
  const float rstd = rsqrtf(var + eps);
  if (lane == 0) { mean_out[row] = mean; rstd_out[row] = rstd; }

  for (int i = lane; i < H; i += QN_WAVE) {
    float f = ld_as_f32(reinterpret_cast<const T*>(xr) + i);
    float wi = ld_as_f32(w + i);
    float bi = ld_as_f32(b + i);
    st_from_f32(yr + i, (f - mean) * rstd * wi + bi);
  }
}

// ---------------------------------------------------------------------------
// backward:
//   xhat = (x - mean) * rstd;  wdy = w * dy
//   dx = (wdy - mean(wdy) - xhat * mean(wdy * xhat)) * rstd
//   dw = sum_rows(dy * xhat);  db = sum_rows(dy)        (fp32 atomics)
// ---------------------------------------------------------------------------
template <typename T, int WPB>
__global__ void layernorm_bwd_kernel(
    const T* __restrict__ dy, const T* __restrict__ x, const T* __restrict__ w,
    const float* __restrict__ mean_in, const float* __restrict__ rstd_in,
    T* __restrict__ dx, float* __restrict__ dw, float* __restrict__ db,
    int rows, int H) {
  const int lane = threadIdx.x & (QN_WAVE - 1);
  const int wave = threadIdx.x / QN_WAVE;
  const long long row = (long long)blockIdx.x * WPB + wave;
  if (row >= rows) return;
  const T* xr = x + row * H;
  const T* dyr = dy + row * H;
  T* dxr = dx + row * H;
  const float mean = mean_in[row], rstd = rstd_in[row];

  float c1 = 0.f, c2 = 0.f;
  for (int i = lane; i < H; i += QN_WAVE) {
    float xf = ld_as_f32(xr + i);
    float dyf = ld_as_f32(dyr + i);
    float wi = ld_as_f32(w + i);
    float xhat = (xf - mean) * rstd;
    float wdy = wi * dyf;
    c1 += wdy;
    c2 += wdy * xhat;
  }
  c1 = wave_reduce_sum(c1) / H;
  c2 = wave_reduce_sum(c2) / H;

  for (int i = lane; i < H; i += QN_WAVE) {
    float xf = ld_as_f32(xr + i);
    float dyf = ld_as_f32(dyr + i);
    float wi = ld_as_f32(w + i);
    float xhat = (xf - mean) * rstd;
    float wdy = wi * dyf;
    st_from_f32(dxr + i, (wdy - c1 - xhat * c2) * rstd);
    atomicAdd(dw + i, dyf * xhat);
    atomicAdd(db + i, dyf);
  }
}

// ---- launchers (called from bindings.cpp) ----------------------------------
template <typename T>
void layernorm_fwd_launch(const T* x, const T* w, const T* b, T* y,
                          float* mean, float* rstd, int rows, int H, float eps,
                          hipStream_t stream) {
  constexpr int WPB = 4;
  dim3 grid((rows + WPB - 1) / WPB);
  hipLaunchKernelGGL((layernorm_fwd_kernel<T, WPB>), grid, dim3(WPB * QN_WAVE), 0,
                     stream, x, w, b, y, mean, rstd, rows, H, eps);
}

template <typename T>
void layernorm_bwd_launch(const T* dy, const T* x, const T* w, const float* mean,
                          const float* rstd, T* dx, float* dw, float* db,
                          int rows, int H, hipStream_t stream) {
  constexpr int WPB = 4;
  dim3 grid((rows + WPB - 1) / WPB);
  hipLaunchKernelGGL((layernorm_bwd_kernel<T, WPB>), grid, dim3(WPB * QN_WAVE), 0,
                     stream, dy, x, w, mean, rstd, dx, dw, db, rows, H);
}

// explicit instantiations
template void layernorm_fwd_launch<float>(const float*, const float*, const float*,
                                          float*, float*, float*, int, int, float, hipStream_t);
template void layernorm_fwd_launch<unsigned short>(const unsigned short*, const unsigned short*,
                                                   const unsigned short*, unsigned short*, float*,
                                                   float*, int, int, float, hipStream_t);
template void layernorm_bwd_launch<float>(const float*, const float*, const float*, const float*,
                                          const float*, float*, float*, float*, int, int, hipStream_t);
template void layernorm_bwd_launch<unsigned short>(const unsigned short*, const unsigned short*,
                                                   const unsigned short*, const float*, const float*,
                                                   unsigned short*, float*, float*, int, int, hipStream_t);
