// Fused LayerNorm fwd/bwd for gfx950 — one wave per row, fp32 accumulate.
// Replaces the implicit eager LayerNorm of the reference.
// Rows are B*T; H ∈ {64, 768, 3072, ...}. bf16 loads vectorized as s16x8
// (guide G13: scalar bf16 ≈2× slower), one pass of sum/sumsq per row.
//
// Backward is two kernels: dx (wave-per-row, no atomics) and a separate
// dw/db column reduction (block-per-column-tile × row-chunks, one
// atomicAdd per column per chunk) — the v1 per-element atomic version
// was 20% of the whole GPT-2 step (rocprof, profiles/r01).
#include "common.h"

// ---------------------------------------------------------------------------
// forward: y = (x - mean) * rstd * w + b ; saves mean, rstd (f32 per row)
// ---------------------------------------------------------------------------
template <typename T, int WPB>
__global__ void layernorm_fwd_kernel(
    const T* __restrict__ x, const T* __restrict__ res,
    const T* __restrict__ w, const T* __restrict__ b,
    T* __restrict__ y, T* __restrict__ sum_out,
    float* __restrict__ mean_out, float* __restrict__ rstd_out,
    int rows, int H, float eps) {
  const int lane = threadIdx.x & (QN_WAVE - 1);
  const int wave = threadIdx.x / QN_WAVE;
  const long long row = (long long)blockIdx.x * WPB + wave;
  if (row >= rows) return;
  const T* xr = x + row * H;
  const T* rr = res ? res + row * H : nullptr;
  T* yr = y + row * H;
  T* sr = sum_out ? sum_out + row * H : nullptr;

  float sum = 0.f, sq = 0.f;
  // cache up to 16 elems/lane in registers (H <= 1024) to avoid re-reads
  float cache[16];
  const bool cached = H <= QN_WAVE * 16;
  const bool vec8 = sizeof(T) == 2 && (H & 7) == 0 && H >= 512;
  if (cached && vec8) {
    // 8-wide bf16 loads (G13); lanes own disjoint 8-elem groups
    int n = 0;
    for (int i = lane * 8; i < H; i += QN_WAVE * 8, n += 8) {
      ld8_f32(reinterpret_cast<const unsigned short*>(xr) + i, cache + n);
      if (rr) {
        float rv[8];
        ld8_f32(reinterpret_cast<const unsigned short*>(rr) + i, rv);
#pragma unroll
        for (int j = 0; j < 8; ++j) cache[n + j] += rv[j];
        if (sr) st8_f32(reinterpret_cast<unsigned short*>(sr) + i, cache + n);
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        sum += cache[n + j];
        sq += cache[n + j] * cache[n + j];
      }
    }
  } else if (cached) {
    int n = 0;
    for (int i = lane; i < H; i += QN_WAVE, ++n) {
      float f = ld_as_f32(xr + i);
      if (rr) {  // fused residual add (SURVEY §2.4 "fused residual-add")
        f += ld_as_f32(rr + i);
        if (sr) st_from_f32(sr + i, f);
      }
      cache[n] = f;
      sum += f; sq += f * f;
    }
  } else {
    for (int i = lane; i < H; i += QN_WAVE) {
      float f = ld_as_f32(xr + i);
      if (rr) {
        f += ld_as_f32(rr + i);
        if (sr) st_from_f32(sr + i, f);
      }
      sum += f; sq += f * f;
    }
  }
  sum = wave_reduce_sum(sum);
  sq = wave_reduce_sum(sq);
  const float mean = sum / H;
  const float var = fmaxf(sq / H - mean * mean, 0.f);
  const float rstd = rsqrtf(var + eps);
  if (lane == 0) { mean_out[row] = mean; rstd_out[row] = rstd; }

  if (cached && vec8) {
    int n = 0;
    for (int i = lane * 8; i < H; i += QN_WAVE * 8, n += 8) {
      float wv[8], bv[8], ov[8];
      ld8_f32(reinterpret_cast<const unsigned short*>(w) + i, wv);
      ld8_f32(reinterpret_cast<const unsigned short*>(b) + i, bv);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        ov[j] = (cache[n + j] - mean) * rstd * wv[j] + bv[j];
      st8_f32(reinterpret_cast<unsigned short*>(yr) + i, ov);
    }
  } else if (cached) {
    int n = 0;
    for (int i = lane; i < H; i += QN_WAVE, ++n) {
      float wi = ld_as_f32(w + i);
      float bi = ld_as_f32(b + i);
      st_from_f32(yr + i, (cache[n] - mean) * rstd * wi + bi);
    }
  } else {
    for (int i = lane; i < H; i += QN_WAVE) {
      float f = ld_as_f32(xr + i);
      if (rr) f += ld_as_f32(rr + i);
      float wi = ld_as_f32(w + i);
      float bi = ld_as_f32(b + i);
      st_from_f32(yr + i, (f - mean) * rstd * wi + bi);
    }
  }
}

// ---------------------------------------------------------------------------
// backward dx (wave per row):
//   xhat = (x - mean) * rstd;  wdy = w * dy
//   dx = (wdy - mean(wdy) - xhat * mean(wdy * xhat)) * rstd
// ---------------------------------------------------------------------------
template <typename T, int WPB>
__global__ void layernorm_bwd_dx_kernel(
    const T* __restrict__ dy, const T* __restrict__ x, const T* __restrict__ w,
    const float* __restrict__ mean_in, const float* __restrict__ rstd_in,
    const T* __restrict__ dsum, T* __restrict__ dx, int rows, int H) {
  const int lane = threadIdx.x & (QN_WAVE - 1);
  const int wave = threadIdx.x / QN_WAVE;
  const long long row = (long long)blockIdx.x * WPB + wave;
  if (row >= rows) return;
  const T* xr = x + row * H;
  const T* dyr = dy + row * H;
  const T* dsr = dsum ? dsum + row * H : nullptr;
  T* dxr = dx + row * H;
  const float mean = mean_in[row], rstd = rstd_in[row];

  float cx[16], cwdy[16];
  const bool cached = H <= QN_WAVE * 16;
  const bool vec8 = sizeof(T) == 2 && (H & 7) == 0 && H >= 512;
  float c1 = 0.f, c2 = 0.f;
  if (cached && vec8) {
    int n = 0;
    for (int i = lane * 8; i < H; i += QN_WAVE * 8, n += 8) {
      float xv[8], wv[8], dv[8];
      ld8_f32(reinterpret_cast<const unsigned short*>(xr) + i, xv);
      ld8_f32(reinterpret_cast<const unsigned short*>(w) + i, wv);
      ld8_f32(reinterpret_cast<const unsigned short*>(dyr) + i, dv);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xhat = (xv[j] - mean) * rstd;
        float wdy = wv[j] * dv[j];
        cx[n + j] = xhat; cwdy[n + j] = wdy;
        c1 += wdy; c2 += wdy * xhat;
      }
    }
  } else if (cached) {
    int n = 0;
    for (int i = lane; i < H; i += QN_WAVE, ++n) {
      float xhat = (ld_as_f32(xr + i) - mean) * rstd;
      float wdy = ld_as_f32(w + i) * ld_as_f32(dyr + i);
      cx[n] = xhat; cwdy[n] = wdy;
      c1 += wdy; c2 += wdy * xhat;
    }
  } else {
    for (int i = lane; i < H; i += QN_WAVE) {
      float xhat = (ld_as_f32(xr + i) - mean) * rstd;
      float wdy = ld_as_f32(w + i) * ld_as_f32(dyr + i);
      c1 += wdy; c2 += wdy * xhat;
    }
  }
  c1 = wave_reduce_sum(c1) / H;
  c2 = wave_reduce_sum(c2) / H;

  if (cached && vec8) {
    int n = 0;
    for (int i = lane * 8; i < H; i += QN_WAVE * 8, n += 8) {
      float ov[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) ov[j] = (cwdy[n + j] - c1 - cx[n + j] * c2) * rstd;
      if (dsr) {
        float dsv[8];
        ld8_f32(reinterpret_cast<const unsigned short*>(dsr) + i, dsv);
#pragma unroll
        for (int j = 0; j < 8; ++j) ov[j] += dsv[j];
      }
      st8_f32(reinterpret_cast<unsigned short*>(dxr) + i, ov);
    }
  } else if (cached) {
    int n = 0;
    for (int i = lane; i < H; i += QN_WAVE, ++n) {
      float v = (cwdy[n] - c1 - cx[n] * c2) * rstd;
      if (dsr) v += ld_as_f32(dsr + i);  // residual-branch grad fused in
      st_from_f32(dxr + i, v);
    }
  } else {
    for (int i = lane; i < H; i += QN_WAVE) {
      float xhat = (ld_as_f32(xr + i) - mean) * rstd;
      float wdy = ld_as_f32(w + i) * ld_as_f32(dyr + i);
      float v = (wdy - c1 - xhat * c2) * rstd;
      if (dsr) v += ld_as_f32(dsr + i);
      st_from_f32(dxr + i, v);
    }
  }
}

// ---------------------------------------------------------------------------
// backward dw/db: column reduction.  grid = (ceil(H/256), ROW_CHUNKS);
// each thread owns one column within its row chunk, one atomicAdd per
// column per chunk.
// ---------------------------------------------------------------------------
template <typename T>
__global__ void layernorm_bwd_dwdb_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const float* __restrict__ mean_in, const float* __restrict__ rstd_in,
    float* __restrict__ dw, float* __restrict__ db, int rows, int H) {
  const int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= H) return;
  const int r0 = blockIdx.y;
  const int nchunks = gridDim.y;
  float aw = 0.f, ab = 0.f;
  for (long long r = r0; r < rows; r += nchunks) {
    float dyv = ld_as_f32(dy + r * H + col);
    float xhat = (ld_as_f32(x + r * H + col) - mean_in[r]) * rstd_in[r];
    aw += dyv * xhat;
    ab += dyv;
  }
  atomicAdd(dw + col, aw);
  atomicAdd(db + col, ab);
}

// ---- launchers -------------------------------------------------------------
template <typename T>
void layernorm_fwd_launch(const T* x, const T* res, const T* w, const T* b,
                          T* y, T* sum_out, float* mean, float* rstd, int rows,
                          int H, float eps, hipStream_t stream) {
  constexpr int WPB = 4;
  dim3 grid((rows + WPB - 1) / WPB);
  hipLaunchKernelGGL((layernorm_fwd_kernel<T, WPB>), grid, dim3(WPB * QN_WAVE), 0,
                     stream, x, res, w, b, y, sum_out, mean, rstd, rows, H, eps);
}

template <typename T>
void layernorm_bwd_launch(const T* dy, const T* x, const T* w, const float* mean,
                          const float* rstd, const T* dsum, T* dx, float* dw,
                          float* db, int rows, int H, hipStream_t stream) {
  constexpr int WPB = 4;
  dim3 grid((rows + WPB - 1) / WPB);
  hipLaunchKernelGGL((layernorm_bwd_dx_kernel<T, WPB>), grid, dim3(WPB * QN_WAVE),
                     0, stream, dy, x, w, mean, rstd, dsum, dx, rows, H);
  int colblocks = (H + 255) / 256;
  int chunks = min(max(rows / 64, 1), 256);
  hipLaunchKernelGGL((layernorm_bwd_dwdb_kernel<T>), dim3(colblocks, chunks),
                     dim3(256), 0, stream, dy, x, mean, rstd, dw, db, rows, H);
}

// explicit instantiations
template void layernorm_fwd_launch<float>(const float*, const float*, const float*, const float*,
                                          float*, float*, float*, float*, int, int, float, hipStream_t);
template void layernorm_fwd_launch<unsigned short>(const unsigned short*, const unsigned short*,
                                                   const unsigned short*, const unsigned short*,
                                                   unsigned short*, unsigned short*, float*,
                                                   float*, int, int, float, hipStream_t);
template void layernorm_bwd_launch<float>(const float*, const float*, const float*, const float*,
                                          const float*, const float*, float*, float*, float*, int, int, hipStream_t);
template void layernorm_bwd_launch<unsigned short>(const unsigned short*, const unsigned short*,
                                                   const unsigned short*, const float*, const float*,
                                                   const unsigned short*, unsigned short*, float*, float*, int, int, hipStream_t);
