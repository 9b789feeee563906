// Common helpers for roc_amd CDNA4 (gfx950) kernels.
// Hand-written HIP for MI355X: wave64, vectorized 16-B lane accesses,
// fp32 accumulation over bf16 storage. No CUDA-compat shims.
#pragma once

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <c10/hip/HIPStream.h>

#define ROC_HIP_CHECK(expr)                                                   \
  do {                                                                        \
    hipError_t _e = (expr);                                                   \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e));      \
  } while (0)

inline hipStream_t roc_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

#define ROC_CHECK_DEV_CONT(t)                                                 \
  TORCH_CHECK((t).is_cuda() && (t).is_contiguous(), #t,                       \
              " must be a contiguous GPU tensor")

// ---------------------------------------------------------------------------
// bf16 <-> f32 bit helpers (bf16 is the top 16 bits of f32)
// ---------------------------------------------------------------------------
__device__ __forceinline__ float bf16_lo(unsigned u) {
  return __uint_as_float(u << 16);
}
__device__ __forceinline__ float bf16_hi(unsigned u) {
  return __uint_as_float(u & 0xffff0000u);
}
// round-to-nearest-even f32 -> bf16 (returned in the low 16 bits)
__device__ __forceinline__ unsigned f32_to_bf16(float f) {
  unsigned x = __float_as_uint(f);
  unsigned rounding = 0x7fffu + ((x >> 16) & 1u);
  x += rounding;
  return x >> 16;
}
__device__ __forceinline__ unsigned pack_bf16(float lo, float hi) {
  return f32_to_bf16(lo) | (f32_to_bf16(hi) << 16);
}

// Element traits: how many elements fit one 16-B lane transaction.
template <typename T>
struct EltTraits;
template <>
struct EltTraits<float> {
  static constexpr int kPerVec = 4;  // float4
};
template <>
struct EltTraits<unsigned short> {  // bf16 storage as raw u16
  static constexpr int kPerVec = 8;  // 8 x bf16 = 16 B
};

// Load EPU elements from a 16B-aligned-or-not location as fp32.
__device__ __forceinline__ void load_f32x4(const float* p, float* acc_dst) {
  const float4 v = *reinterpret_cast<const float4*>(p);
  acc_dst[0] = v.x; acc_dst[1] = v.y; acc_dst[2] = v.z; acc_dst[3] = v.w;
}
__device__ __forceinline__ void load_bf16x8(const unsigned short* p,
                                            float* acc_dst) {
  const uint4 v = *reinterpret_cast<const uint4*>(p);
  acc_dst[0] = bf16_lo(v.x); acc_dst[1] = bf16_hi(v.x);
  acc_dst[2] = bf16_lo(v.y); acc_dst[3] = bf16_hi(v.y);
  acc_dst[4] = bf16_lo(v.z); acc_dst[5] = bf16_hi(v.z);
  acc_dst[6] = bf16_lo(v.w); acc_dst[7] = bf16_hi(v.w);
}
__device__ __forceinline__ void store_f32x4(float* p, const float* a) {
  *reinterpret_cast<float4*>(p) = make_float4(a[0], a[1], a[2], a[3]);
}
__device__ __forceinline__ void store_bf16x8(unsigned short* p,
                                             const float* a) {
  uint4 v;
  v.x = pack_bf16(a[0], a[1]);
  v.y = pack_bf16(a[2], a[3]);
  v.z = pack_bf16(a[4], a[5]);
  v.w = pack_bf16(a[6], a[7]);
  *reinterpret_cast<uint4*>(p) = v;
}
__device__ __forceinline__ void store_f32x8(float* p, const float* a) {
  store_f32x4(p, a);
  store_f32x4(p + 4, a + 4);
}

__device__ __forceinline__ float elt_to_f32(float x) { return x; }
__device__ __forceinline__ float elt_to_f32(unsigned short x) {
  return __uint_as_float(((unsigned)x) << 16);
}
__device__ __forceinline__ void f32_to_elt(float f, float* d) { *d = f; }
__device__ __forceinline__ void f32_to_elt(float f, unsigned short* d) {
  *d = (unsigned short)f32_to_bf16(f);
}

constexpr int kBlock = 256;  // 4 wavefronts

inline int roc_grid_1d(long long work, int per_block, int cap = 4096) {
  long long g = (work + per_block - 1) / per_block;
  if (g < 1) g = 1;
  if (g > cap) g = cap;
  return (int)g;
}
