// Fused softmax + cross-entropy gradient + metrics (CDNA4).
//
// dlogits[i,c] = (softmax(logits[i,:C])[c] - onehot) * [mask[i]==Train] * scale
// metrics[8]  = {roc_loss_sum, ce_loss_sum,
//                train_correct, train_total, val_correct, val_total,
//                test_correct, test_total}
//
// The class dim may be stored PADDED (stride >= C, e.g. 41 classes in 48
// columns for 16-B-aligned bf16 rows): softmax runs over the true C,
// pad columns get zero gradient.
//
// Fast path (C <= 64): ONE global read per row — each lane holds one
// class in a register for max/sum/grad. Metrics accumulate in wave
// registers across all rows, then one LDS reduction per block and 8
// global atomics per block (the naive per-row atomic version serialized
// ~700k adds onto 8 words and cost 7.2 ms/epoch on Reddit — see
// profiles/r01_baseline_kernel_stats.md).
// Replaces reference cudnnSoftmaxForward + softmax_backward + calc_loss
// (`softmax_kernel.cu:19-171`).

#include "common.h"

namespace {

constexpr int kMaskTrain = 1;
constexpr float kNegInf = -3.4e38f;

template <typename T, bool NARROW>
__global__ __launch_bounds__(kBlock) void softmax_ce_kernel(
    T* __restrict__ dl, float* __restrict__ metrics,
    const T* __restrict__ logits, const int64_t* __restrict__ labels,
    const int* __restrict__ mask, float grad_scale, int num_rows, int C,
    int stride) {
  __shared__ float macc[8];
  if (threadIdx.x < 8) macc[threadIdx.x] = 0.f;
  __syncthreads();

  const int wave = (blockIdx.x * blockDim.x + threadIdx.x) / 64;
  const int lane = threadIdx.x & 63;
  const int nwaves = (gridDim.x * blockDim.x) / 64;

  // wave-local metric accumulators (identical on all lanes after reduce)
  float acc[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) acc[j] = 0.f;

  for (int row = wave; row < num_rows; row += nwaves) {
    const T* lrow = logits + (int64_t)row * stride;
    const int label = (int)labels[row];
    const int mval = mask[row];
    const float train = (mval == kMaskTrain) ? grad_scale : 0.f;
    float p_true, best;
    int best_c;

    if (NARROW) {
      const float v = (lane < C) ? elt_to_f32(lrow[lane]) : kNegInf;
      float mx = v;
#pragma unroll
      for (int w = 32; w >= 1; w >>= 1) mx = fmaxf(mx, __shfl_xor(mx, w));
      const float e = (lane < C) ? __expf(v - mx) : 0.f;
      float sum = e;
      best = v;
      best_c = (lane < C) ? lane : C;
#pragma unroll
      for (int w = 32; w >= 1; w >>= 1) {
        sum += __shfl_xor(sum, w);
        const float ob = __shfl_xor(best, w);
        const int oc = __shfl_xor(best_c, w);
        if (ob > best || (ob == best && oc < best_c)) { best = ob; best_c = oc; }
      }
      const float inv_sum = 1.f / sum;
      const float p = e * inv_sum;
      if (lane < stride) {
        T* drow = dl + (int64_t)row * stride;
        const float g = (lane < C)
            ? (p - (lane == label ? 1.f : 0.f)) * train : 0.f;
        f32_to_elt(g, drow + lane);
      }
      p_true = __shfl(p, label);
    } else {
      float mx = kNegInf;
      for (int c = lane; c < C; c += 64) mx = fmaxf(mx, elt_to_f32(lrow[c]));
#pragma unroll
      for (int w = 32; w >= 1; w >>= 1) mx = fmaxf(mx, __shfl_xor(mx, w));
      float sum = 0.f;
      best = kNegInf;
      best_c = C;
      for (int c = lane; c < C; c += 64) {
        const float v = elt_to_f32(lrow[c]);
        sum += __expf(v - mx);
        if (v > best || (v == best && c < best_c)) { best = v; best_c = c; }
      }
#pragma unroll
      for (int w = 32; w >= 1; w >>= 1) {
        sum += __shfl_xor(sum, w);
        const float ob = __shfl_xor(best, w);
        const int oc = __shfl_xor(best_c, w);
        if (ob > best || (ob == best && oc < best_c)) { best = ob; best_c = oc; }
      }
      const float inv_sum = 1.f / sum;
      T* drow = dl + (int64_t)row * stride;
      for (int c = lane; c < stride; c += 64) {
        const float g = (c < C)
            ? (__expf(elt_to_f32(lrow[c]) - mx) * inv_sum -
               (c == label ? 1.f : 0.f)) * train
            : 0.f;
        f32_to_elt(g, drow + c);
      }
      p_true = __expf(elt_to_f32(lrow[label]) - mx) * inv_sum;
    }

    if (mval == kMaskTrain) {
      acc[0] += 1.f - p_true;
      acc[1] += -__logf(fmaxf(p_true, 1e-12f));
    }
    if (mval >= 1 && mval <= 3) {
      if (best_c == label) acc[2 * mval] += 1.f;
      acc[2 * mval + 1] += 1.f;
    }
  }

  // block reduction: one lane per wave -> LDS -> 8 global atomics
  if (lane == 0) {
#pragma unroll
    for (int j = 0; j < 8; ++j) atomicAdd(&macc[j], acc[j]);
  }
  __syncthreads();
  if (threadIdx.x < 8 && macc[threadIdx.x] != 0.f)
    atomicAdd(&metrics[threadIdx.x], macc[threadIdx.x]);
}

}  // namespace

void softmax_ce(torch::Tensor dl, torch::Tensor metrics, torch::Tensor logits,
                torch::Tensor labels, torch::Tensor mask, double grad_scale,
                int64_t num_classes) {
  ROC_CHECK_DEV_CONT(dl);
  ROC_CHECK_DEV_CONT(metrics);
  ROC_CHECK_DEV_CONT(logits);
  ROC_CHECK_DEV_CONT(labels);
  ROC_CHECK_DEV_CONT(mask);
  TORCH_CHECK(labels.scalar_type() == torch::kInt64);
  TORCH_CHECK(mask.scalar_type() == torch::kInt32);
  TORCH_CHECK(metrics.scalar_type() == torch::kFloat32 && metrics.numel() == 8);
  const int n = (int)logits.size(0);
  const int stride = (int)logits.size(1);
  const int C = num_classes > 0 ? (int)num_classes : stride;
  TORCH_CHECK(C <= stride, "num_classes exceeds logits width");
  const int waves_per_block = kBlock / 64;
  const int grid = roc_grid_1d(n, waves_per_block, 2048);
  auto s = roc_stream();
#define ROC_SCE(T, NARROW, PT)                                            \
  hipLaunchKernelGGL((softmax_ce_kernel<T, NARROW>), dim3(grid),          \
                     dim3(kBlock), 0, s, (T*)dl.data_ptr(),               \
                     metrics.data_ptr<float>(), (const T*)logits.data_ptr(), \
                     labels.data_ptr<int64_t>(), mask.data_ptr<int>(),    \
                     (float)grad_scale, n, C, stride)
  if (logits.scalar_type() == torch::kBFloat16) {
    if (C <= 64 && stride <= 64) ROC_SCE(unsigned short, true, u16);
    else ROC_SCE(unsigned short, false, u16);
  } else if (logits.scalar_type() == torch::kFloat32) {
    if (C <= 64 && stride <= 64) ROC_SCE(float, true, f32);
    else ROC_SCE(float, false, f32);
  } else {
    TORCH_CHECK(false, "softmax_ce: unsupported dtype");
  }
#undef ROC_SCE
  ROC_HIP_CHECK(hipGetLastError());
}
