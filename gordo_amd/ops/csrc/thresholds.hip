// Device threshold/statistics kernels for MI355X (gfx950) —
// SURVEY.md §2.3 K10 (rolling-window thresholds), K11 (smoothing),
// K12 (quantile threshold).
//
// The reference computes DiffBased thresholds as
// `series.rolling(w).min().max()` (gordo diff.py:229-254), smoothed
// thresholds with the same trailing-window statistic over `window`,
// and KFCV thresholds as `rolling(window).median().quantile(p)`
// (diff.py:566-635). These kernels batch that math over a whole pack
// ([G] machines x [N] timesteps x [F] tags) so the CV scoring phase
// consumes device fold predictions directly instead of a per-machine
// pandas loop.
//
// Semantics match pandas exactly at fp32:
//  * trail_min_max: windows with fewer than w elements are excluded
//    (pandas NaN-skipping max over the valid suffix).
//  * windowed_quantile (the rolling median with q=0.5): min_periods=w
//    — ANY NaN inside a window yields NaN for that position.
//  * row_quantile: NaNs dropped, linear interpolation between order
//    statistics (pandas/numpy default).
#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <cmath>
#include <vector>

#define DEV_INLINE __device__ __forceinline__

// ---------------------------------------------------------------------------
// K10: out[r] = max_{i >= w-1} min(X[r, i-w+1 .. i])
// One 256-thread workgroup per row; each thread scans its strided
// output positions with an O(w) window min (N*w per row is ~0.6M ops
// at the fleet shape — microseconds), then a block max-reduce.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void trail_min_max_kernel(
    const float* __restrict__ X, float* __restrict__ out, int N, int w) {
  const int r = blockIdx.x;
  const float* row = X + (size_t)r * N;
  const int tid = threadIdx.x;
  float best = -INFINITY;
  bool any = false;
  for (int i = w - 1 + tid; i < N; i += 256) {
    float m = row[i];
    for (int k = i - w + 1; k < i; ++k) m = fminf(m, row[k]);
    // NaN windows are skipped like pandas' NaN-skipping max
    if (!isnan(m)) { best = fmaxf(best, m); any = true; }
  }
  __shared__ float red[256];
  __shared__ int cnt[256];
  red[tid] = best;
  cnt[tid] = any ? 1 : 0;
  __syncthreads();
  for (int s = 128; s > 0; s >>= 1) {
    if (tid < s) {
      red[tid] = fmaxf(red[tid], red[tid + s]);
      cnt[tid] += cnt[tid + s];
    }
    __syncthreads();
  }
  if (tid == 0) out[r] = cnt[0] ? red[0] : NAN;
}

// ---------------------------------------------------------------------------
// Shared helper: bitonic sort of LDS buffer of P (power of two) floats
// with 256 threads, ascending; NaN/pad slots pre-filled with +inf.
// ---------------------------------------------------------------------------
DEV_INLINE void bitonic_sort_lds(float* buf, int P) {
  const int tid = threadIdx.x;
  for (int k = 2; k <= P; k <<= 1) {
    for (int j = k >> 1; j > 0; j >>= 1) {
      for (int idx = tid; idx < P; idx += 256) {
        int ixj = idx ^ j;
        if (ixj > idx) {
          float a = buf[idx], b = buf[ixj];
          bool up = ((idx & k) == 0);
          if ((a > b) == up) { buf[idx] = b; buf[ixj] = a; }
        }
      }
      __syncthreads();
    }
  }
}

DEV_INLINE float interp_quantile(const float* sorted, int n, float q) {
  // numpy/pandas 'linear': pos = q*(n-1)
  if (n <= 0) return NAN;
  float pos = q * (n - 1);
  int lo = (int)floorf(pos);
  int hi = lo + 1 < n ? lo + 1 : lo;
  float frac = pos - lo;
  return sorted[lo] + (sorted[hi] - sorted[lo]) * frac;
}

// ---------------------------------------------------------------------------
// K11: rolling-window quantile (q=0.5 == pandas rolling median,
// min_periods=w). One workgroup per output element (r, i); window
// padded to P slots of +inf in LDS, bitonic-sorted.
// out[r, i] for i in [0, N-w]; grid.x = R * (N-w+1).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void windowed_quantile_kernel(
    const float* __restrict__ X, float* __restrict__ out, int N, int w,
    int P, float q) {
  extern __shared__ float buf[];
  const int no = N - w + 1;
  const int r = blockIdx.x / no;
  const int i0 = blockIdx.x % no;
  const float* row = X + (size_t)r * N + i0;
  const int tid = threadIdx.x;
  __shared__ int has_nan;
  if (tid == 0) has_nan = 0;
  __syncthreads();
  for (int k = tid; k < P; k += 256) {
    float v = (k < w) ? row[k] : INFINITY;
    if (isnan(v)) { has_nan = 1; v = INFINITY; }
    buf[k] = v;
  }
  __syncthreads();
  if (has_nan) {  // pandas min_periods=w: any NaN in window -> NaN
    if (tid == 0) out[(size_t)r * no + i0] = NAN;
    return;
  }
  bitonic_sort_lds(buf, P);
  if (tid == 0) out[(size_t)r * no + i0] = interp_quantile(buf, w, q);
}

// ---------------------------------------------------------------------------
// K12: per-row quantile with NaN dropping + linear interpolation.
// One workgroup per row; row padded to P (<= 16384) slots in LDS.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void row_quantile_kernel(
    const float* __restrict__ X, float* __restrict__ out, int N, int P,
    float q) {
  extern __shared__ float buf[];
  const int r = blockIdx.x;
  const float* row = X + (size_t)r * N;
  const int tid = threadIdx.x;
  __shared__ int nan_count[256];
  int my_nan = 0;
  for (int k = tid; k < P; k += 256) {
    float v = (k < N) ? row[k] : INFINITY;
    if (isnan(v)) { v = INFINITY; my_nan++; }
    buf[k] = v;
  }
  nan_count[tid] = my_nan;
  __syncthreads();
  for (int s = 128; s > 0; s >>= 1) {
    if (tid < s) nan_count[tid] += nan_count[tid + s];
    __syncthreads();
  }
  const int n_valid = N - nan_count[0];
  bitonic_sort_lds(buf, P);
  if (tid == 0) out[r] = interp_quantile(buf, n_valid, q);
}

// ---------------------------------------------------------------------------
namespace gordo_thresholds {

static inline int next_pow2(int n) {
  int p = 1;
  while (p < n) p <<= 1;
  return p;
}

torch::Tensor trail_min_max(torch::Tensor X, int64_t w) {
  TORCH_CHECK(X.is_cuda() && X.dim() == 2, "X must be [R, N] on GPU");
  auto xc = X.to(torch::kFloat32).contiguous();
  int R = xc.size(0), N = xc.size(1);
  auto out = torch::empty({R}, xc.options());
  if (N < w) {
    out.fill_(std::numeric_limits<float>::quiet_NaN());
    return out;
  }
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  hipLaunchKernelGGL(trail_min_max_kernel, dim3(R), dim3(256), 0, stream,
                     xc.data_ptr<float>(), out.data_ptr<float>(), N,
                     (int)w);
  return out;
}

torch::Tensor windowed_quantile(torch::Tensor X, int64_t w, double q) {
  TORCH_CHECK(X.is_cuda() && X.dim() == 2, "X must be [R, N] on GPU");
  auto xc = X.to(torch::kFloat32).contiguous();
  int R = xc.size(0), N = xc.size(1);
  TORCH_CHECK(N >= w, "window longer than series");
  int P = next_pow2((int)w);
  TORCH_CHECK(P <= 8192, "window too large for LDS sort");
  int no = N - (int)w + 1;
  auto out = torch::empty({R, no}, xc.options());
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  hipLaunchKernelGGL(windowed_quantile_kernel, dim3((unsigned)R * no),
                     dim3(256), P * sizeof(float), stream,
                     xc.data_ptr<float>(), out.data_ptr<float>(), N,
                     (int)w, P, (float)q);
  return out;
}

torch::Tensor row_quantile(torch::Tensor X, double q) {
  TORCH_CHECK(X.is_cuda() && X.dim() == 2, "X must be [R, N] on GPU");
  auto xc = X.to(torch::kFloat32).contiguous();
  int R = xc.size(0), N = xc.size(1);
  int P = next_pow2(N);
  TORCH_CHECK(P <= 16384, "series too large for LDS sort (chunk it)");
  auto out = torch::empty({R}, xc.options());
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  hipLaunchKernelGGL(row_quantile_kernel, dim3(R), dim3(256),
                     P * sizeof(float), stream, xc.data_ptr<float>(),
                     out.data_ptr<float>(), N, P, (float)q);
  return out;
}

}  // namespace gordo_thresholds
