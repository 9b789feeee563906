// Counter-based Philox4x32-10 dropout (CDNA4).
//
// Stateless: mask is a pure function of (seed, call offset, element index),
// so the backward pass regenerates the identical mask instead of reading a
// stored one — zero mask memory, zero extra HBM traffic (the reference
// keeps persistent cuDNN reserve state per op, `dropout_kernel.cu:44-56`).

#include "common.h"

namespace {

__device__ __forceinline__ uint2 philox_round(uint2 ctr01, uint2 ctr23,
                                              uint2 key, uint2* out23) {
  // one round of philox4x32
  const unsigned M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
  unsigned hi0 = __umulhi(M0, ctr01.x), lo0 = M0 * ctr01.x;
  unsigned hi1 = __umulhi(M1, ctr23.x), lo1 = M1 * ctr23.x;
  uint2 r01 = make_uint2(hi1 ^ ctr01.y ^ key.x, lo1);
  uint2 r23 = make_uint2(hi0 ^ ctr23.y ^ key.y, lo0);
  *out23 = r23;
  return r01;
}

__device__ __forceinline__ uint4 philox4x32_10(unsigned long long idx,
                                               unsigned offset,
                                               unsigned long long seed) {
  uint2 key = make_uint2((unsigned)seed, (unsigned)(seed >> 32));
  uint2 c01 = make_uint2((unsigned)idx, (unsigned)(idx >> 32));
  uint2 c23 = make_uint2(offset, 0x9E3779B9u);
  const unsigned W0 = 0x9E3779B9u, W1 = 0xBB67AE85u;
#pragma unroll
  for (int r = 0; r < 10; ++r) {
    uint2 n23;
    uint2 n01 = philox_round(c01, c23, key, &n23);
    c01 = n01;
    c23 = n23;
    key.x += W0;
    key.y += W1;
  }
  return make_uint4(c01.x, c01.y, c23.x, c23.y);
}

constexpr float kU32ToUnit = 2.3283064365386963e-10f;  // 2^-32

// bf16 fast path: 8 elements per thread = one 16-B load/store and two
// Philox draws (indices 2u, 2u+1 — the SAME mask stream as the scalar
// path, which covers 4 elements per draw). The scalar path's 2-B
// loads/stores were the bottleneck (ALU is trivial next to them).
__global__ __launch_bounds__(kBlock) void dropout_kernel_bf16x8(
    unsigned short* __restrict__ out, const unsigned short* __restrict__ x,
    float p, float inv_keep, unsigned long long seed, unsigned offset,
    const long long* __restrict__ counter, int64_t units8) {
  const unsigned off_eff =
      counter ? (unsigned)(*counter) * 65536u + offset : offset;
  for (int64_t u = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       u < units8; u += (int64_t)gridDim.x * blockDim.x) {
    const uint4 ra = philox4x32_10((unsigned long long)(2 * u), off_eff,
                                   seed);
    const uint4 rb = philox4x32_10((unsigned long long)(2 * u + 1),
                                   off_eff, seed);
    const unsigned rnd[8] = {ra.x, ra.y, ra.z, ra.w,
                             rb.x, rb.y, rb.z, rb.w};
    float v[8];
    load_bf16x8(x + u * 8, v);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      v[j] *= (rnd[j] * kU32ToUnit >= p) ? inv_keep : 0.f;
    store_bf16x8(out + u * 8, v);
  }
}

template <typename T>
__global__ __launch_bounds__(kBlock) void dropout_kernel(
    T* __restrict__ out, const T* __restrict__ x, float p, float inv_keep,
    unsigned long long seed, unsigned offset,
    const long long* __restrict__ counter, int64_t n) {
  // hipGraph-replay support: the epoch index lives on the DEVICE so a
  // captured graph draws fresh Philox streams on every replay.
  const unsigned off_eff =
      counter ? (unsigned)(*counter) * 65536u + offset : offset;
  const int64_t units = (n + 3) / 4;
  for (int64_t u = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; u < units;
       u += (int64_t)gridDim.x * blockDim.x) {
    const uint4 r = philox4x32_10((unsigned long long)u, off_eff, seed);
    const unsigned rnd[4] = {r.x, r.y, r.z, r.w};
    const int64_t i0 = u * 4;
    const int cnt = (int)min((int64_t)4, n - i0);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      if (j < cnt) {
        const float keep = (rnd[j] * kU32ToUnit >= p) ? inv_keep : 0.f;
        f32_to_elt(elt_to_f32(x[i0 + j]) * keep, out + i0 + j);
      }
    }
  }
}

}  // namespace

void dropout_fwd(torch::Tensor out, torch::Tensor x, double p, int64_t seed,
                 int64_t offset, c10::optional<torch::Tensor> counter) {
  ROC_CHECK_DEV_CONT(out);
  ROC_CHECK_DEV_CONT(x);
  TORCH_CHECK(out.scalar_type() == x.scalar_type());
  TORCH_CHECK(p >= 0.0 && p < 1.0, "dropout p out of range");
  const int64_t n = x.numel();
  const float inv_keep = (float)(1.0 / (1.0 - p));
  const int grid = roc_grid_1d((n + 3) / 4, kBlock, 2048);
  const long long* cnt = nullptr;
  if (counter.has_value()) {
    TORCH_CHECK(counter->scalar_type() == torch::kInt64 && counter->is_cuda());
    cnt = (const long long*)counter->data_ptr<int64_t>();
  }
  auto s = roc_stream();
  if (x.scalar_type() == torch::kBFloat16 && n % 8 == 0) {
    const int grid8 = roc_grid_1d(n / 8, kBlock, 2048);
    hipLaunchKernelGGL(dropout_kernel_bf16x8, dim3(grid8), dim3(kBlock), 0,
                       s, (unsigned short*)out.data_ptr(),
                       (const unsigned short*)x.data_ptr(), (float)p,
                       inv_keep, (unsigned long long)seed, (unsigned)offset,
                       cnt, n / 8);
  } else if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL((dropout_kernel<unsigned short>), dim3(grid),
                       dim3(kBlock), 0, s, (unsigned short*)out.data_ptr(),
                       (const unsigned short*)x.data_ptr(), (float)p, inv_keep,
                       (unsigned long long)seed, (unsigned)offset, cnt, n);
  } else if (x.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL((dropout_kernel<float>), dim3(grid), dim3(kBlock), 0, s,
                       out.data_ptr<float>(), x.data_ptr<float>(), (float)p,
                       inv_keep, (unsigned long long)seed, (unsigned)offset,
                       cnt, n);
  } else {
    TORCH_CHECK(false, "dropout: unsupported dtype");
  }
  ROC_HIP_CHECK(hipGetLastError());
}
