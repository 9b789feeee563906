// Elementwise + rowwise-scale kernels (CDNA4, 16-B vectorized lanes).
// Covers the reference's graphnorm (`graphnorm_kernel.cu`), activation
// (`activation_kernel.cu`), element add/mul (`element_kernel.cu`) and
// relu-backward (`linear_kernel.cu:120-127`) ops, fp32 math over
// bf16/f32 storage.

#include "common.h"

namespace {

// ---- generic vectorized 1-arg / 2-arg maps --------------------------------

template <typename T, typename Op>
__global__ __launch_bounds__(kBlock) void map1_kernel(
    T* __restrict__ out, const T* __restrict__ a, int64_t n, Op op) {
  constexpr int EPU = EltTraits<T>::kPerVec;
  const int64_t units = (n + EPU - 1) / EPU;
  for (int64_t u = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; u < units;
       u += (int64_t)gridDim.x * blockDim.x) {
    const int64_t i0 = u * EPU;
    float v[EPU];
    if (i0 + EPU <= n) {
      if constexpr (EPU == 8) load_bf16x8(a + i0, v); else load_f32x4(a + i0, v);
#pragma unroll
      for (int j = 0; j < EPU; ++j) v[j] = op(v[j], i0 + j);
      if constexpr (EPU == 8) store_bf16x8(out + i0, v);
      else store_f32x4(out + i0, v);
    } else {
      for (int64_t i = i0; i < n; ++i)
        f32_to_elt(op(elt_to_f32(a[i]), i), out + i);
    }
  }
}

template <typename T, typename Op>
__global__ __launch_bounds__(kBlock) void map2_kernel(
    T* __restrict__ out, const T* __restrict__ a, const T* __restrict__ b,
    int64_t n, Op op) {
  constexpr int EPU = EltTraits<T>::kPerVec;
  const int64_t units = (n + EPU - 1) / EPU;
  for (int64_t u = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; u < units;
       u += (int64_t)gridDim.x * blockDim.x) {
    const int64_t i0 = u * EPU;
    float va[EPU], vb[EPU];
    if (i0 + EPU <= n) {
      if constexpr (EPU == 8) { load_bf16x8(a + i0, va); load_bf16x8(b + i0, vb); }
      else                    { load_f32x4(a + i0, va);  load_f32x4(b + i0, vb); }
#pragma unroll
      for (int j = 0; j < EPU; ++j) va[j] = op(va[j], vb[j]);
      if constexpr (EPU == 8) store_bf16x8(out + i0, va);
      else store_f32x4(out + i0, va);
    } else {
      for (int64_t i = i0; i < n; ++i)
        f32_to_elt(op(elt_to_f32(a[i]), elt_to_f32(b[i])), out + i);
    }
  }
}

struct OpRelu { __device__ float operator()(float x, int64_t) const { return x > 0.f ? x : 0.f; } };
struct OpSigmoid { __device__ float operator()(float x, int64_t) const { return 1.f / (1.f + __expf(-x)); } };
struct OpAdd { __device__ float operator()(float a, float b) const { return a + b; } };
struct OpMul { __device__ float operator()(float a, float b) const { return a * b; } };
struct OpReluBwd { __device__ float operator()(float dy, float y) const { return y > 0.f ? dy : 0.f; } };
struct OpSigmoidBwd { __device__ float operator()(float dy, float y) const { return dy * y * (1.f - y); } };

struct OpRowScale {
  const float* scale;
  int64_t D;
  __device__ float operator()(float x, int64_t i) const {
    return x * scale[i / D];
  }
};

template <typename T, typename Op>
void launch_map1(T* out, const T* a, int64_t n, Op op, hipStream_t s) {
  constexpr int EPU = EltTraits<T>::kPerVec;
  const int grid = roc_grid_1d((n + EPU - 1) / EPU, kBlock, 2048);
  hipLaunchKernelGGL((map1_kernel<T, Op>), dim3(grid), dim3(kBlock), 0, s,
                     out, a, n, op);
}
template <typename T, typename Op>
void launch_map2(T* out, const T* a, const T* b, int64_t n, Op op,
                 hipStream_t s) {
  constexpr int EPU = EltTraits<T>::kPerVec;
  const int grid = roc_grid_1d((n + EPU - 1) / EPU, kBlock, 2048);
  hipLaunchKernelGGL((map2_kernel<T, Op>), dim3(grid), dim3(kBlock), 0, s,
                     out, a, b, n, op);
}

template <typename Op>
void dispatch_map1(torch::Tensor out, torch::Tensor a, Op op) {
  ROC_CHECK_DEV_CONT(out);
  ROC_CHECK_DEV_CONT(a);
  TORCH_CHECK(out.sizes() == a.sizes() && out.scalar_type() == a.scalar_type());
  auto s = roc_stream();
  const int64_t n = a.numel();
  if (a.scalar_type() == torch::kBFloat16) {
    launch_map1((unsigned short*)out.data_ptr(),
                (const unsigned short*)a.data_ptr(), n, op, s);
  } else if (a.scalar_type() == torch::kFloat32) {
    launch_map1(out.data_ptr<float>(), a.data_ptr<float>(), n, op, s);
  } else {
    TORCH_CHECK(false, "unsupported dtype");
  }
  ROC_HIP_CHECK(hipGetLastError());
}

template <typename Op>
void dispatch_map2(torch::Tensor out, torch::Tensor a, torch::Tensor b, Op op) {
  ROC_CHECK_DEV_CONT(out);
  ROC_CHECK_DEV_CONT(a);
  ROC_CHECK_DEV_CONT(b);
  TORCH_CHECK(a.sizes() == b.sizes() && a.scalar_type() == b.scalar_type());
  auto s = roc_stream();
  const int64_t n = a.numel();
  if (a.scalar_type() == torch::kBFloat16) {
    launch_map2((unsigned short*)out.data_ptr(),
                (const unsigned short*)a.data_ptr(),
                (const unsigned short*)b.data_ptr(), n, op, s);
  } else if (a.scalar_type() == torch::kFloat32) {
    launch_map2(out.data_ptr<float>(), a.data_ptr<float>(),
                b.data_ptr<float>(), n, op, s);
  } else {
    TORCH_CHECK(false, "unsupported dtype");
  }
  ROC_HIP_CHECK(hipGetLastError());
}

}  // namespace

void rowscale(torch::Tensor out, torch::Tensor x, torch::Tensor scale) {
  ROC_CHECK_DEV_CONT(scale);
  TORCH_CHECK(scale.scalar_type() == torch::kFloat32, "scale must be fp32");
  TORCH_CHECK(x.dim() == 2 && scale.size(0) == x.size(0));
  OpRowScale op{scale.data_ptr<float>(), x.size(1)};
  dispatch_map1(out, x, op);
}

void relu_fwd(torch::Tensor out, torch::Tensor x) {
  dispatch_map1(out, x, OpRelu{});
}
void sigmoid_fwd(torch::Tensor out, torch::Tensor x) {
  dispatch_map1(out, x, OpSigmoid{});
}
void relu_bwd(torch::Tensor dx, torch::Tensor dy, torch::Tensor y) {
  dispatch_map2(dx, dy, y, OpReluBwd{});
}
void sigmoid_bwd(torch::Tensor dx, torch::Tensor dy, torch::Tensor y) {
  dispatch_map2(dx, dy, y, OpSigmoidBwd{});
}
void ewise_add(torch::Tensor out, torch::Tensor a, torch::Tensor b) {
  dispatch_map2(out, a, b, OpAdd{});
}
void ewise_mul(torch::Tensor out, torch::Tensor a, torch::Tensor b) {
  dispatch_map2(out, a, b, OpMul{});
}

// fp32 -> bf16 cast with optional per-row scale, one pass (the strip-
// blocked SpMM epilogue: fp32 strip accumulator -> scaled bf16 out;
// replaces a torch mul + a torch cast = 3 passes over the data).
// Requires D % 8 == 0 (engine pads feature dims to 8).
namespace {
__global__ __launch_bounds__(kBlock) void cast_rowscale_kernel(
    unsigned short* __restrict__ out, const float* __restrict__ x,
    const float* __restrict__ scale, int64_t D, int64_t units) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t u = blockIdx.x * blockDim.x + threadIdx.x; u < units;
       u += stride) {
    const int64_t base = u * 8;
    float a[8];
    load_f32x4(x + base, a);
    load_f32x4(x + base + 4, a + 4);
    if (scale) {
      const float s = scale[base / D];
#pragma unroll
      for (int j = 0; j < 8; ++j) a[j] *= s;
    }
    store_bf16x8(out + base, a);
  }
}
}  // namespace

void cast_rowscale(torch::Tensor out, torch::Tensor x,
                   c10::optional<torch::Tensor> scale) {
  ROC_CHECK_DEV_CONT(out);
  ROC_CHECK_DEV_CONT(x);
  TORCH_CHECK(out.scalar_type() == torch::kBFloat16, "out must be bf16");
  TORCH_CHECK(x.scalar_type() == torch::kFloat32, "x must be fp32");
  TORCH_CHECK(out.sizes() == x.sizes(), "shape mismatch");
  TORCH_CHECK(x.dim() == 2 && x.size(1) % 8 == 0,
              "cast_rowscale needs D %% 8 == 0");
  const float* sp = nullptr;
  if (scale.has_value()) {
    ROC_CHECK_DEV_CONT(*scale);
    TORCH_CHECK(scale->scalar_type() == torch::kFloat32 &&
                scale->numel() == x.size(0), "bad scale");
    sp = scale->data_ptr<float>();
  }
  const int64_t units = x.numel() / 8;
  const int grid = roc_grid_1d(units, kBlock, 2048);
  hipLaunchKernelGGL(cast_rowscale_kernel, dim3(grid), dim3(kBlock), 0,
                     roc_stream(), (unsigned short*)out.data_ptr(),
                     x.data_ptr<float>(), sp, x.size(1), units);
  ROC_HIP_CHECK(hipGetLastError());
}
