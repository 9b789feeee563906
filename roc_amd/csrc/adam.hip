// Fused Adam with L2-coupled weight decay (CDNA4, fp32, float4 lanes).
// Semantics of reference `optimizer_kernel.cu:43-63`:
//   gt = g + wd*w;  m = b1*m+(1-b1)*gt;  v = b2*v+(1-b2)*gt^2;
//   w -= alpha_t * m / (sqrt(v)+eps)
// alpha_t carries the bias correction (computed host-side,
// `optimizer.cc:79-85`). The reference first summed per-partition grad
// replicas on ONE GPU; here the grad arrives already RCCL-all-reduced.

#include "common.h"

namespace {

__global__ __launch_bounds__(kBlock) void adam_kernel(
    float* __restrict__ w, const float* __restrict__ g,
    float* __restrict__ m, float* __restrict__ v, float alpha, float b1,
    float b2, float eps, float wd,
    const long long* __restrict__ step, float decay_rate, int decay_steps,
    int64_t n) {
  // hipGraph-replay support: with `step` given, `alpha` is the BASE lr and
  // the bias-corrected, decayed step size is derived on device from the
  // step counter (semantics of optimizer.cc:79-85 + gnn.cc:100-101).
  if (step) {
    const float t = (float)(*step);
    const float lr_t =
        alpha * __powf(decay_rate, floorf(t / (float)decay_steps));
    alpha = lr_t * sqrtf(1.f - __powf(b2, t)) / (1.f - __powf(b1, t));
  }
  const int64_t units = (n + 3) / 4;
  for (int64_t u = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; u < units;
       u += (int64_t)gridDim.x * blockDim.x) {
    const int64_t i0 = u * 4;
    if (i0 + 4 <= n) {
      float4 wv = *reinterpret_cast<float4*>(w + i0);
      const float4 gv = *reinterpret_cast<const float4*>(g + i0);
      float4 mv = *reinterpret_cast<float4*>(m + i0);
      float4 vv = *reinterpret_cast<float4*>(v + i0);
      float* pw = &wv.x;
      const float* pg = &gv.x;
      float* pm = &mv.x;
      float* pv = &vv.x;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const float gt = pg[j] + wd * pw[j];
        pm[j] = b1 * pm[j] + (1.f - b1) * gt;
        pv[j] = b2 * pv[j] + (1.f - b2) * gt * gt;
        pw[j] -= alpha * pm[j] / (sqrtf(pv[j]) + eps);
      }
      *reinterpret_cast<float4*>(w + i0) = wv;
      *reinterpret_cast<float4*>(m + i0) = mv;
      *reinterpret_cast<float4*>(v + i0) = vv;
    } else {
      for (int64_t i = i0; i < n; ++i) {
        const float gt = g[i] + wd * w[i];
        m[i] = b1 * m[i] + (1.f - b1) * gt;
        v[i] = b2 * v[i] + (1.f - b2) * gt * gt;
        w[i] -= alpha * m[i] / (sqrtf(v[i]) + eps);
      }
    }
  }
}

}  // namespace

void adam_step(torch::Tensor w, torch::Tensor g, torch::Tensor m,
               torch::Tensor v, double alpha, double b1, double b2, double eps,
               double wd, c10::optional<torch::Tensor> step,
               double decay_rate, int64_t decay_steps) {
  ROC_CHECK_DEV_CONT(w);
  ROC_CHECK_DEV_CONT(g);
  ROC_CHECK_DEV_CONT(m);
  ROC_CHECK_DEV_CONT(v);
  TORCH_CHECK(w.scalar_type() == torch::kFloat32 &&
              g.scalar_type() == torch::kFloat32, "adam: fp32 masters only");
  const int64_t n = w.numel();
  TORCH_CHECK(g.numel() == n && m.numel() == n && v.numel() == n);
  const int grid = roc_grid_1d((n + 3) / 4, kBlock, 1024);
  const long long* st = nullptr;
  if (step.has_value()) {
    TORCH_CHECK(step->scalar_type() == torch::kInt64 && step->is_cuda());
    st = (const long long*)step->data_ptr<int64_t>();
  }
  hipLaunchKernelGGL(adam_kernel, dim3(grid), dim3(kBlock), 0, roc_stream(),
                     w.data_ptr<float>(), g.data_ptr<float>(),
                     m.data_ptr<float>(), v.data_ptr<float>(), (float)alpha,
                     (float)b1, (float)b2, (float)eps, (float)wd, st,
                     (float)decay_rate, (int)decay_steps, n);
  ROC_HIP_CHECK(hipGetLastError());
}
