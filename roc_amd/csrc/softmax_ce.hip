// Fused softmax + cross-entropy gradient + metrics (CDNA4, one wave/row).
//
// dlogits[i,c] = (softmax(logits[i])[c] - onehot) * [mask[i]==Train] * scale
// metrics[8]  = {roc_loss_sum, ce_loss_sum,
//                train_correct, train_total, val_correct, val_total,
//                test_correct, test_total}   (float32 atomics)
//
// Replaces the reference's cudnnSoftmaxForward + softmax_backward +
// calc_loss trio (`softmax_kernel.cu:19-171`) with a single pass; the
// reference's "loss" Σ(1-p_true) is kept as roc_loss, true CE is also
// accumulated. Math in fp32 from bf16/f32 logits.

#include "common.h"

namespace {

constexpr int kMaskTrain = 1;

template <typename T>
__global__ __launch_bounds__(kBlock) void softmax_ce_kernel(
    T* __restrict__ dl, float* __restrict__ metrics,
    const T* __restrict__ logits, const int64_t* __restrict__ labels,
    const int* __restrict__ mask, float grad_scale, int num_rows, int C) {
  const int wave = (blockIdx.x * blockDim.x + threadIdx.x) / 64;
  const int lane = threadIdx.x & 63;
  const int nwaves = (gridDim.x * blockDim.x) / 64;

  for (int row = wave; row < num_rows; row += nwaves) {
    const T* lrow = logits + (int64_t)row * C;
    // 1) wave max
    float mx = -3.4e38f;
    for (int c = lane; c < C; c += 64) mx = fmaxf(mx, elt_to_f32(lrow[c]));
#pragma unroll
    for (int w = 32; w >= 1; w >>= 1) mx = fmaxf(mx, __shfl_xor(mx, w));
    // 2) wave sum of exp + argmax
    float sum = 0.f;
    float best = -3.4e38f;
    int best_c = C;
    for (int c = lane; c < C; c += 64) {
      const float v = elt_to_f32(lrow[c]);
      sum += __expf(v - mx);
      if (v > best || (v == best && c < best_c)) {
        best = v;
        best_c = c;
      }
    }
#pragma unroll
    for (int w = 32; w >= 1; w >>= 1) {
      sum += __shfl_xor(sum, w);
      const float ob = __shfl_xor(best, w);
      const int oc = __shfl_xor(best_c, w);
      if (ob > best || (ob == best && oc < best_c)) {
        best = ob;
        best_c = oc;
      }
    }
    const float inv_sum = 1.f / sum;
    const int label = (int)labels[row];
    const int mval = mask[row];
    const float train = (mval == kMaskTrain) ? grad_scale : 0.f;
    // 3) gradient
    T* drow = dl + (int64_t)row * C;
    for (int c = lane; c < C; c += 64) {
      const float p = __expf(elt_to_f32(lrow[c]) - mx) * inv_sum;
      f32_to_elt((p - (c == label ? 1.f : 0.f)) * train, drow + c);
    }
    // 4) metrics (lane 0)
    if (lane == 0) {
      const float p_true = __expf(elt_to_f32(lrow[label]) - mx) * inv_sum;
      if (mval == kMaskTrain) {
        atomicAdd(&metrics[0], 1.f - p_true);
        atomicAdd(&metrics[1], -__logf(fmaxf(p_true, 1e-12f)));
      }
      if (mval >= 1 && mval <= 3) {
        if (best_c == label) atomicAdd(&metrics[2 * mval], 1.f);
        atomicAdd(&metrics[2 * mval + 1], 1.f);
      }
    }
  }
}

}  // namespace

void softmax_ce(torch::Tensor dl, torch::Tensor metrics, torch::Tensor logits,
                torch::Tensor labels, torch::Tensor mask, double grad_scale) {
  ROC_CHECK_DEV_CONT(dl);
  ROC_CHECK_DEV_CONT(metrics);
  ROC_CHECK_DEV_CONT(logits);
  ROC_CHECK_DEV_CONT(labels);
  ROC_CHECK_DEV_CONT(mask);
  TORCH_CHECK(labels.scalar_type() == torch::kInt64);
  TORCH_CHECK(mask.scalar_type() == torch::kInt32);
  TORCH_CHECK(metrics.scalar_type() == torch::kFloat32 && metrics.numel() == 8);
  const int n = (int)logits.size(0);
  const int C = (int)logits.size(1);
  const int waves_per_block = kBlock / 64;
  const int grid = roc_grid_1d(n, waves_per_block, 4096);
  auto s = roc_stream();
  if (logits.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL((softmax_ce_kernel<unsigned short>), dim3(grid),
                       dim3(kBlock), 0, s, (unsigned short*)dl.data_ptr(),
                       metrics.data_ptr<float>(),
                       (const unsigned short*)logits.data_ptr(),
                       labels.data_ptr<int64_t>(), mask.data_ptr<int>(),
                       (float)grad_scale, n, C);
  } else if (logits.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL((softmax_ce_kernel<float>), dim3(grid), dim3(kBlock), 0,
                       s, dl.data_ptr<float>(), metrics.data_ptr<float>(),
                       logits.data_ptr<float>(), labels.data_ptr<int64_t>(),
                       mask.data_ptr<int>(), (float)grad_scale, n, C);
  } else {
    TORCH_CHECK(false, "softmax_ce: unsupported dtype");
  }
  ROC_HIP_CHECK(hipGetLastError());
}
