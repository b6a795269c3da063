// Fused anomaly-scoring kernel — the serving hot
// path of DiffBasedAnomalyDetector.anomaly (reference diff.py:310-458):
// one pass over (model_output, y) emitting every response column family:
//   tag_scaled[n,f]   = |(out*scale+min) - (y*scale+min)|
//   total_scaled[n]   = mean_f tag_scaled^2
//   tag_unscaled[n,f] = |out - y|
//   total_unscaled[n] = mean_f tag_unscaled^2
//   confidence[n,f]   = tag_unscaled / feature_threshold[f]
//   total_conf[n]     = total_scaled / aggregate_threshold
// One workgroup per row-block; per-row reductions via wave shuffles.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <vector>

__global__ void anomaly_score_kernel(
    const float* __restrict__ out, const float* __restrict__ y,
    const float* __restrict__ scale, const float* __restrict__ minv,
    const float* __restrict__ feat_thr, float agg_thr,
    float* __restrict__ tag_scaled, float* __restrict__ total_scaled,
    float* __restrict__ tag_unscaled, float* __restrict__ total_unscaled,
    float* __restrict__ confidence, float* __restrict__ total_conf,
    int N, int F, int has_thr) {
  int n = blockIdx.x * (blockDim.x / 64) + (threadIdx.x >> 6);
  int lane = threadIdx.x & 63;
  if (n >= N) return;
  const float* orow = out + (size_t)n * F;
  const float* yrow = y + (size_t)n * F;
  float ss = 0.f, su = 0.f;
  for (int f = lane; f < F; f += 64) {
    float d = orow[f] - yrow[f];
    float du = fabsf(d);
    float ds = fabsf(d * scale[f]);  // (a*s+m)-(b*s+m) == (a-b)*s
    tag_scaled[(size_t)n * F + f] = ds;
    tag_unscaled[(size_t)n * F + f] = du;
    if (has_thr) confidence[(size_t)n * F + f] = du / feat_thr[f];
    ss += ds * ds;
    su += du * du;
  }
  #pragma unroll
  for (int off = 32; off; off >>= 1) {
    ss += __shfl_down(ss, off, 64);
    su += __shfl_down(su, off, 64);
  }
  if (lane == 0) {
    float ts = ss / F;
    total_scaled[n] = ts;
    total_unscaled[n] = su / F;
    if (has_thr) total_conf[n] = ts / agg_thr;
  }
}

namespace gordo_anomaly {

std::vector<torch::Tensor> anomaly_score(
    torch::Tensor out, torch::Tensor y, torch::Tensor scale,
    torch::Tensor minv, c10::optional<torch::Tensor> feat_thr,
    double agg_thr) {
  TORCH_CHECK(out.is_cuda() && out.dim() == 2, "out must be [N,F] on GPU");
  auto oc = out.to(torch::kFloat32).contiguous();
  auto yc = y.to(torch::kFloat32).contiguous();
  auto sc = scale.to(torch::kFloat32).contiguous();
  auto mc = minv.to(torch::kFloat32).contiguous();
  int N = oc.size(0), F = oc.size(1);
  auto opts = oc.options();
  auto tag_scaled = torch::empty({N, F}, opts);
  auto total_scaled = torch::empty({N}, opts);
  auto tag_unscaled = torch::empty({N, F}, opts);
  auto total_unscaled = torch::empty({N}, opts);
  int has_thr = feat_thr.has_value() ? 1 : 0;
  auto conf = torch::empty({has_thr ? N : 0, F}, opts);
  auto total_conf = torch::empty({has_thr ? N : 0}, opts);
  const float* thr_ptr = nullptr;
  torch::Tensor thr_c;
  if (has_thr) {
    thr_c = feat_thr->to(torch::kFloat32).contiguous().to(oc.device());
    thr_ptr = thr_c.data_ptr<float>();
  }
  int rows_per_block = 4;  // 256 threads = 4 waves, one row per wave
  int blocks = (N + rows_per_block - 1) / rows_per_block;
  hipLaunchKernelGGL(anomaly_score_kernel, dim3(blocks), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream().stream(),
                     oc.data_ptr<float>(), yc.data_ptr<float>(),
                     sc.data_ptr<float>(), mc.data_ptr<float>(), thr_ptr,
                     (float)agg_thr, tag_scaled.data_ptr<float>(),
                     total_scaled.data_ptr<float>(),
                     tag_unscaled.data_ptr<float>(),
                     total_unscaled.data_ptr<float>(),
                     conf.data_ptr<float>(), total_conf.data_ptr<float>(),
                     N, F, has_thr);
  return {tag_scaled, total_scaled, tag_unscaled, total_unscaled, conf,
          total_conf};
}

}  // namespace gordo_anomaly
