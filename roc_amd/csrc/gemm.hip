// Hand-written MFMA GEMMs for the linear op (CDNA4 / gfx950).
//
//  gemm_rr : C[M,N] = A[M,K] @ B[K,N]   (+ fused ReLU / rowwise scale)
//            A row-major, B passed PRE-TRANSPOSED as Bt[N,K] so both LDS
//            stages are direct coalesced copies (no transpose writes).
//            Replaces reference cublasSgemm fwd/dX (`linear_kernel.cu:76,227`).
//  gemm_atb: C[Ka,N] += A[R,Ka]^T @ B[R,N]   (fp32 out, split-K atomics)
//            the weight-gradient GEMM (reference `linear_kernel.cu:220`,
//            beta=1 accumulate); reduction dim R is the node count (~10^5-6),
//            so blocks split R and atomically accumulate fp32 partials.
//
// Element types: bf16 (mfma_f32_16x16x32_bf16, the flagship path) and
// exact fp32 (mfma_f32_16x16x4f32 at the f32 vector rate — gfx950 has no
// xf32; this keeps the reference's fp32-only mode available on GPU).
// Shapes are tall-skinny (M ~ nodes, K/N ~ 41..640): 128-row M-tiles,
// K-step 32, fp32 accumulation, zero-filled staging for every edge tail,
// double-buffered LDS with the T14 split (loads issued before the
// previous tile's MFMAs, LDS writes after).

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

namespace {

constexpr int BM = 128;   // M rows per block
constexpr int BK = 32;    // K per step
constexpr int PADK = 40;  // LDS row stride in ELEMENTS (16-B aligned)

template <typename ET>
struct GemmTraits;
template <>
struct GemmTraits<unsigned short> {  // bf16 storage
  static constexpr int kEPS = 8;     // elements per 16-B segment
  using Seg = short8;
  static __device__ __forceinline__ Seg zero() {
    return Seg{0, 0, 0, 0, 0, 0, 0, 0};
  }
};
template <>
struct GemmTraits<float> {
  static constexpr int kEPS = 4;
  using Seg = float4;
  static __device__ __forceinline__ Seg zero() {
    return make_float4(0.f, 0.f, 0.f, 0.f);
  }
};

// ---- staging: LOAD half (global -> regs) + WRITE half (regs -> LDS) ----
template <typename ET, int ROWS, bool ALIGNED>
__device__ __forceinline__ void stage_load(
    typename GemmTraits<ET>::Seg (
        &regs)[ROWS * (BK / GemmTraits<ET>::kEPS) / kBlock],
    const ET* __restrict__ g, int row0, int nrows, int64_t ld, int k0,
    int K) {
  constexpr int EPS = GemmTraits<ET>::kEPS;
  constexpr int SPR = BK / EPS;               // segments per row
  constexpr int ROWS_PER_PASS = kBlock / SPR;
  const int seg = threadIdx.x % SPR;
  const int r_in = threadIdx.x / SPR;
#pragma unroll
  for (int pass = 0; pass < ROWS / ROWS_PER_PASS; ++pass) {
    const int gr = row0 + pass * ROWS_PER_PASS + r_in;
    const int gk = k0 + seg * EPS;
    auto v = GemmTraits<ET>::zero();
    if (gr < nrows) {
      const ET* p = g + (int64_t)gr * ld + gk;
      if (ALIGNED && gk + EPS <= K) {
        v = *reinterpret_cast<const typename GemmTraits<ET>::Seg*>(p);
      } else {
        const int nv = min(EPS, K - gk);
        for (int j = 0; j < nv; ++j) v[j] = p[j];
      }
    }
    regs[pass] = v;
  }
}

template <typename ET, int ROWS>
__device__ __forceinline__ void stage_write(
    ET* __restrict__ lds,
    const typename GemmTraits<ET>::Seg (
        &regs)[ROWS * (BK / GemmTraits<ET>::kEPS) / kBlock]) {
  constexpr int EPS = GemmTraits<ET>::kEPS;
  constexpr int SPR = BK / EPS;
  constexpr int ROWS_PER_PASS = kBlock / SPR;
  const int seg = threadIdx.x % SPR;
  const int r_in = threadIdx.x / SPR;
#pragma unroll
  for (int pass = 0; pass < ROWS / ROWS_PER_PASS; ++pass) {
    const int r = pass * ROWS_PER_PASS + r_in;
    *reinterpret_cast<typename GemmTraits<ET>::Seg*>(
        &lds[r * PADK + seg * EPS]) = regs[pass];
  }
}

// one 16x16 output fragment's worth of MFMAs for a BK=32 K-step.
// bf16: a single mfma_f32_16x16x32_bf16; fp32: 8 x mfma_f32_16x16x4f32.
template <typename ET>
__device__ __forceinline__ f32x4 frag_mfma(const ET* a_lds, const ET* b_lds,
                                           int mrow, int nrow, int khalf,
                                           f32x4 acc) {
  if constexpr (sizeof(ET) == 2) {
    const short8 af = *reinterpret_cast<const short8*>(
        &a_lds[mrow * PADK + khalf * 8]);
    const short8 bf = *reinterpret_cast<const short8*>(
        &b_lds[nrow * PADK + khalf * 8]);
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc, 0, 0, 0);
  } else {
#pragma unroll
    for (int kk = 0; kk < BK / 4; ++kk) {
      const float af = a_lds[mrow * PADK + kk * 4 + khalf];
      const float bf = b_lds[nrow * PADK + kk * 4 + khalf];
      acc = __builtin_amdgcn_mfma_f32_16x16x4f32(af, bf, acc, 0, 0, 0);
    }
    return acc;
  }
}

// ---------------------------------------------------------------------------
// gemm_rr
// ---------------------------------------------------------------------------

template <typename ET, int BN, bool RELU, bool ALIGNED_A, bool ALIGNED_B>
__global__ __launch_bounds__(kBlock) void gemm_rr_kernel(
    ET* __restrict__ C, const ET* __restrict__ A, const ET* __restrict__ Bt,
    const float* __restrict__ row_scale, int M, int N, int K) {
  constexpr int NFRAG = BN / 16;
  __shared__ ET a_lds[2][BM * PADK];
  __shared__ ET b_lds[2][BN * PADK];

  const int m_blk = blockIdx.x * BM;
  const int n_blk = blockIdx.y * BN;
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  const int l15 = lane & 15;
  const int khalf = lane >> 4;
  const int m_wave = wave * 32;

  f32x4 acc[2][NFRAG];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < NFRAG; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  typename GemmTraits<ET>::Seg ra[BM * (BK / GemmTraits<ET>::kEPS) / kBlock];
  typename GemmTraits<ET>::Seg rb[BN * (BK / GemmTraits<ET>::kEPS) / kBlock];
  const int nk = (K + BK - 1) / BK;
  stage_load<ET, BM, ALIGNED_A>(ra, A, m_blk, M, K, 0, K);
  stage_load<ET, BN, ALIGNED_B>(rb, Bt, n_blk, N, K, 0, K);
  stage_write<ET, BM>(a_lds[0], ra);
  stage_write<ET, BN>(b_lds[0], rb);
  __syncthreads();
  int cur = 0;
  for (int kt = 0; kt < nk; ++kt) {
    if (kt + 1 < nk) {  // issue next tile's loads before the MFMAs (T14)
      stage_load<ET, BM, ALIGNED_A>(ra, A, m_blk, M, K, (kt + 1) * BK, K);
      stage_load<ET, BN, ALIGNED_B>(rb, Bt, n_blk, N, K, (kt + 1) * BK, K);
    }
#pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
      const int mrow = m_wave + mi * 16 + l15;
#pragma unroll
      for (int ni = 0; ni < NFRAG; ++ni) {
        acc[mi][ni] = frag_mfma<ET>(a_lds[cur], b_lds[cur], mrow,
                                    ni * 16 + l15, khalf, acc[mi][ni]);
      }
    }
    if (kt + 1 < nk) {
      stage_write<ET, BM>(a_lds[cur ^ 1], ra);
      stage_write<ET, BN>(b_lds[cur ^ 1], rb);
      __syncthreads();
      cur ^= 1;
    }
  }

  // epilogue: C[row][col], row = m_frag + (lane>>4)*4 + j, col = n_frag + l15
#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
    for (int ni = 0; ni < NFRAG; ++ni) {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int row = m_blk + m_wave + mi * 16 + khalf * 4 + j;
        const int col = n_blk + ni * 16 + l15;
        if (row < M && col < N) {
          float v = acc[mi][ni][j];
          if (row_scale) v *= row_scale[row];  // fused rowwise scale
          if (RELU) v = v > 0.f ? v : 0.f;
          f32_to_elt(v, &C[(int64_t)row * N + col]);
        }
      }
    }
  }
}

template <typename ET, int BN>
void launch_rr(ET* C, const ET* A, const ET* Bt, const float* row_scale,
               int M, int N, int K, bool relu, hipStream_t s) {
  dim3 grid((M + BM - 1) / BM, (N + BN - 1) / BN);
  const bool al = (K % GemmTraits<ET>::kEPS) == 0;  // row stride = K
#define ROC_RR_CASE(RELU_, ALA)                                               \
  hipLaunchKernelGGL((gemm_rr_kernel<ET, BN, RELU_, ALA, ALA>), grid,         \
                     dim3(kBlock), 0, s, C, A, Bt, row_scale, M, N, K)
  if (relu) { if (al) ROC_RR_CASE(true, true); else ROC_RR_CASE(true, false); }
  else      { if (al) ROC_RR_CASE(false, true); else ROC_RR_CASE(false, false); }
#undef ROC_RR_CASE
}

// ---------------------------------------------------------------------------
// gemm_atb (split-K weight-grad GEMM, fp32 atomic accumulate)
// ---------------------------------------------------------------------------

constexpr int BKA = 64;  // Ka tile (output rows)
constexpr int BNW = 64;  // N tile (output cols)
constexpr int RB = 32;   // reduction rows per step

// stage a [RB x 64] tile TRANSPOSED into LDS[64][PADK] (lds[c][r]).
//
// bf16 swizzle: the plain [c][r] image puts every lane of a half-wave
// on the same bank pair (column stride 64*PADK bytes for a fixed r is
// 0 mod the 32 write banks), measured at 22% of gemm_atb_wide's wave
// cycles in SQ_LDS_BANK_CONFLICT (profiles/r29). Storing row
// r ^ ((cseg&3)<<3) spreads the 8 column-segment lanes over 4 bank
// groups while keeping each khalf-read's 8 elements contiguous; the
// reader compensates with khalf ^ ((c>>3)&3) (frag_mfma_tswz).
// LOAD half: globals -> regs (issue-only; the s_waitcnt lands in the
// WRITE half, so callers can put a full MFMA phase between the two —
// the T14 split; the fused version had zero latency hiding)
template <typename ET, bool ALIGNED>
__device__ __forceinline__ void stage_loadT(
    typename GemmTraits<ET>::Seg (&regs)[RB * 64 / kBlock /
                                         GemmTraits<ET>::kEPS],
    const ET* __restrict__ g, int r0, int nrows, int64_t ld, int c0,
    int ncols) {
  constexpr int EPS = GemmTraits<ET>::kEPS;
  constexpr int SPR = 64 / EPS;           // segments per row of 64 cols
  constexpr int RPP = kBlock / SPR;       // rows per pass
  const int cseg = threadIdx.x % SPR;
  const int r_in = threadIdx.x / SPR;
#pragma unroll
  for (int pass = 0; pass < RB / RPP; ++pass) {
    const int r = pass * RPP + r_in;
    const int gr = r0 + r;
    auto v = GemmTraits<ET>::zero();
    if (gr < nrows) {
      const ET* p = g + (int64_t)gr * ld + c0 + cseg * EPS;
      const int nv = min(EPS, ncols - (c0 + cseg * EPS));
      if (ALIGNED && nv >= EPS) {
        v = *reinterpret_cast<const typename GemmTraits<ET>::Seg*>(p);
      } else {
        for (int j = 0; j < max(nv, 0); ++j) v[j] = p[j];
      }
    }
    regs[pass] = v;
  }
}

// WRITE half: regs -> transposed (bf16: bank-swizzled) LDS image
template <typename ET>
__device__ __forceinline__ void stage_writeT(
    ET* __restrict__ lds,
    const typename GemmTraits<ET>::Seg (&regs)[RB * 64 / kBlock /
                                               GemmTraits<ET>::kEPS]) {
  constexpr int EPS = GemmTraits<ET>::kEPS;
  constexpr int SPR = 64 / EPS;
  constexpr int RPP = kBlock / SPR;
  const int cseg = threadIdx.x % SPR;
  const int r_in = threadIdx.x / SPR;
#pragma unroll
  for (int pass = 0; pass < RB / RPP; ++pass) {
    const int r = pass * RPP + r_in;
    const int rs = (sizeof(ET) == 2) ? (r ^ ((cseg & 3) << 3)) : r;
#pragma unroll
    for (int j = 0; j < EPS; ++j)
      lds[(cseg * EPS + j) * PADK + rs] = (ET)regs[pass][j];
  }
}

template <typename ET, bool ALIGNED>
__device__ __forceinline__ void stage_tile_T(
    ET* __restrict__ lds, const ET* __restrict__ g, int r0, int nrows,
    int64_t ld, int c0, int ncols) {
  typename GemmTraits<ET>::Seg regs[RB * 64 / kBlock /
                                    GemmTraits<ET>::kEPS];
  stage_loadT<ET, ALIGNED>(regs, g, r0, nrows, ld, c0, ncols);
  stage_writeT<ET>(lds, regs);
}

// fragment reader for the swizzled transposed images (bf16; the fp32
// image is stored unswizzled and read by the generic frag_mfma)
template <typename ET>
__device__ __forceinline__ f32x4 frag_mfma_tswz(
    const ET* a_lds, const ET* b_lds, int mrow, int nrow, int khalf,
    f32x4 acc) {
  if constexpr (sizeof(ET) == 2) {
    const int ka = (khalf ^ ((mrow >> 3) & 3)) * 8;
    const int kb = (khalf ^ ((nrow >> 3) & 3)) * 8;
    const short8 af =
        *reinterpret_cast<const short8*>(&a_lds[mrow * PADK + ka]);
    const short8 bf =
        *reinterpret_cast<const short8*>(&b_lds[nrow * PADK + kb]);
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc, 0, 0, 0);
  } else {
    return frag_mfma<ET>(a_lds, b_lds, mrow, nrow, khalf, acc);
  }
}

template <typename ET, bool ALIGNED_A, bool ALIGNED_B>
__global__ __launch_bounds__(kBlock) void gemm_atb_kernel(
    float* __restrict__ C, const ET* __restrict__ A, const ET* __restrict__ B,
    int R, int Ka, int N, int rows_per_split) {
  __shared__ ET at_lds[BKA * PADK];
  __shared__ ET bt_lds[BNW * PADK];

  const int i_blk = blockIdx.x * BKA;
  const int n_blk = blockIdx.y * BNW;
  const int r_begin = blockIdx.z * rows_per_split;
  const int r_end = min(R, r_begin + rows_per_split);

  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  const int l15 = lane & 15;
  const int khalf = lane >> 4;
  const int i_wave = wave * 16;  // 16 Ka rows per wave

  f32x4 acc[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) acc[j] = {0.f, 0.f, 0.f, 0.f};

  // T14 split pipeline: chunk r0+RB's global loads are issued right
  // after the barrier, a full MFMA phase before their s_waitcnt in
  // stage_writeT (the fused stage had zero latency hiding)
  typename GemmTraits<ET>::Seg ra[RB * 64 / kBlock / GemmTraits<ET>::kEPS];
  typename GemmTraits<ET>::Seg rb[RB * 64 / kBlock / GemmTraits<ET>::kEPS];
  stage_loadT<ET, ALIGNED_A>(ra, A, r_begin, r_end, Ka, i_blk, Ka);
  stage_loadT<ET, ALIGNED_B>(rb, B, r_begin, r_end, N, n_blk, N);
  for (int r0 = r_begin; r0 < r_end; r0 += RB) {
    stage_writeT<ET>(at_lds, ra);
    stage_writeT<ET>(bt_lds, rb);
    __syncthreads();
    if (r0 + RB < r_end) {
      stage_loadT<ET, ALIGNED_A>(ra, A, r0 + RB, r_end, Ka, i_blk, Ka);
      stage_loadT<ET, ALIGNED_B>(rb, B, r0 + RB, r_end, N, n_blk, N);
    }
    const int irow = i_wave + l15;
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      acc[ni] = frag_mfma_tswz<ET>(at_lds, bt_lds, irow, ni * 16 + l15,
                                   khalf, acc[ni]);
    }
    __syncthreads();
  }

#pragma unroll
  for (int ni = 0; ni < 4; ++ni) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int i = i_blk + i_wave + khalf * 4 + j;
      const int n = n_blk + ni * 16 + l15;
      if (i < Ka && n < N && acc[ni][j] != 0.f)
        atomicAdd(&C[(int64_t)i * N + n], acc[ni][j]);
    }
  }
}

// 128x128 output-tile variant: 4x fewer cross-tile re-reads of A and B
// (the 64x64 kernel re-read A per N-tile and B per Ka-tile — at
// [608,256] that is 2.3 GB/call vs 1.16 GB here; measured r2c20).
// Each wave owns 2 Ka row-groups (mi) x 8 N fragments (ni).
template <typename ET, bool ALIGNED_A, bool ALIGNED_B>
__global__ __launch_bounds__(kBlock) void gemm_atb_wide_kernel(
    float* __restrict__ C, const ET* __restrict__ A, const ET* __restrict__ B,
    int R, int Ka, int N, int rows_per_split) {
  __shared__ ET at_lds[2 * BKA * PADK];
  __shared__ ET bt_lds[2 * BNW * PADK];

  const int i_blk = blockIdx.x * (2 * BKA);
  const int n_blk = blockIdx.y * (2 * BNW);
  const int r_begin = blockIdx.z * rows_per_split;
  const int r_end = min(R, r_begin + rows_per_split);

  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  const int l15 = lane & 15;
  const int khalf = lane >> 4;
  const int i_wave = wave * 16;

  f32x4 acc[2][8];
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int ni = 0; ni < 8; ++ni) acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

  // T14 split pipeline (see gemm_atb_kernel): 4 tile loads in flight
  // across each MFMA phase
  typename GemmTraits<ET>::Seg ra0[RB * 64 / kBlock / GemmTraits<ET>::kEPS];
  typename GemmTraits<ET>::Seg ra1[RB * 64 / kBlock / GemmTraits<ET>::kEPS];
  typename GemmTraits<ET>::Seg rb0[RB * 64 / kBlock / GemmTraits<ET>::kEPS];
  typename GemmTraits<ET>::Seg rb1[RB * 64 / kBlock / GemmTraits<ET>::kEPS];
  stage_loadT<ET, ALIGNED_A>(ra0, A, r_begin, r_end, Ka, i_blk, Ka);
  stage_loadT<ET, ALIGNED_A>(ra1, A, r_begin, r_end, Ka, i_blk + BKA, Ka);
  stage_loadT<ET, ALIGNED_B>(rb0, B, r_begin, r_end, N, n_blk, N);
  stage_loadT<ET, ALIGNED_B>(rb1, B, r_begin, r_end, N, n_blk + BNW, N);
  for (int r0 = r_begin; r0 < r_end; r0 += RB) {
    stage_writeT<ET>(at_lds, ra0);
    stage_writeT<ET>(at_lds + BKA * PADK, ra1);
    stage_writeT<ET>(bt_lds, rb0);
    stage_writeT<ET>(bt_lds + BNW * PADK, rb1);
    __syncthreads();
    if (r0 + RB < r_end) {
      stage_loadT<ET, ALIGNED_A>(ra0, A, r0 + RB, r_end, Ka, i_blk, Ka);
      stage_loadT<ET, ALIGNED_A>(ra1, A, r0 + RB, r_end, Ka, i_blk + BKA,
                                 Ka);
      stage_loadT<ET, ALIGNED_B>(rb0, B, r0 + RB, r_end, N, n_blk, N);
      stage_loadT<ET, ALIGNED_B>(rb1, B, r0 + RB, r_end, N, n_blk + BNW,
                                 N);
    }
    const int irow = i_wave + l15;
#pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
      for (int ni = 0; ni < 8; ++ni) {
        acc[mi][ni] = frag_mfma_tswz<ET>(at_lds + mi * BKA * PADK, bt_lds,
                                         irow, ni * 16 + l15, khalf,
                                         acc[mi][ni]);
      }
    }
    __syncthreads();
  }

#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 8; ++ni) {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int i = i_blk + mi * BKA + i_wave + khalf * 4 + j;
        const int n = n_blk + ni * 16 + l15;
        if (i < Ka && n < N && acc[mi][ni][j] != 0.f)
          atomicAdd(&C[(int64_t)i * N + n], acc[mi][ni][j]);
      }
    }
  }
}

template <typename ET>
void launch_atb(float* c, const ET* a, const ET* b, int R, int Ka, int N,
                hipStream_t s) {
  const bool wide = Ka >= 2 * BKA && N >= 2 * BNW;
  const int tw = wide ? 2 : 1;
  const int tiles = ((Ka + tw * BKA - 1) / (tw * BKA)) *
                    ((N + tw * BNW - 1) / (tw * BNW));
  // split-K count trades fill (tiles*splitk workgroups) against the
  // atomic RMW volume into C (Ka*N*splitk adds); pinned once at first
  // launch (same hipGraph-capture rule as the SpMM knobs)
  static const int splitk_env = [] {
    const char* e = getenv("ROC_ATB_SPLITK");
    return e ? atoi(e) : 0;
  }();
  int splitk = splitk_env > 0 ? splitk_env : 2048 / max(tiles, 1);
  splitk = max(1, min(splitk, (R + RB - 1) / RB));
  const char* det = getenv("ROC_DETERMINISTIC");
  if (det && det[0] == '1') splitk = 1;  // bit-reproducible dW (slower)
  int rows_per_split = ((R + splitk - 1) / splitk + RB - 1) / RB * RB;
  splitk = (R + rows_per_split - 1) / rows_per_split;
  dim3 grid((Ka + tw * BKA - 1) / (tw * BKA),
            (N + tw * BNW - 1) / (tw * BNW), splitk);
  const bool ala = (Ka % GemmTraits<ET>::kEPS) == 0;
  const bool alb = (N % GemmTraits<ET>::kEPS) == 0;
#define ROC_ATB_CASE(ALA, ALB)                                                \
  do {                                                                        \
    if (wide) {                                                               \
      hipLaunchKernelGGL((gemm_atb_wide_kernel<ET, ALA, ALB>), grid,          \
                         dim3(kBlock), 0, s, c, a, b, R, Ka, N,               \
                         rows_per_split);                                     \
    } else {                                                                  \
      hipLaunchKernelGGL((gemm_atb_kernel<ET, ALA, ALB>), grid,               \
                         dim3(kBlock), 0, s, c, a, b, R, Ka, N,               \
                         rows_per_split);                                     \
    }                                                                         \
  } while (0)
  if (ala) { if (alb) ROC_ATB_CASE(true, true); else ROC_ATB_CASE(true, false); }
  else     { if (alb) ROC_ATB_CASE(false, true); else ROC_ATB_CASE(false, false); }
#undef ROC_ATB_CASE
}

}  // namespace

void gemm_rr(torch::Tensor C, torch::Tensor A, torch::Tensor Bt, bool relu,
             c10::optional<torch::Tensor> row_scale) {
  ROC_CHECK_DEV_CONT(C);
  ROC_CHECK_DEV_CONT(A);
  ROC_CHECK_DEV_CONT(Bt);
  TORCH_CHECK(A.scalar_type() == Bt.scalar_type() &&
                  A.scalar_type() == C.scalar_type(),
              "gemm_rr: mixed dtypes");
  const int M = (int)A.size(0), K = (int)A.size(1);
  const int N = (int)Bt.size(0);
  TORCH_CHECK(Bt.size(1) == K, "gemm_rr: Bt must be [N,K]");
  TORCH_CHECK(C.size(0) == M && C.size(1) == N, "gemm_rr: C shape");
  if (M == 0 || N == 0) return;  // empty partition
  if (K == 0) { C.zero_(); return; }
  auto s = roc_stream();
  const float* rs = nullptr;
  if (row_scale.has_value()) {
    TORCH_CHECK(row_scale->scalar_type() == torch::kFloat32 &&
                    row_scale->numel() == M,
                "row_scale must be fp32 [M]");
    rs = row_scale->data_ptr<float>();
  }
  if (A.scalar_type() == torch::kBFloat16) {
    auto* c = (unsigned short*)C.data_ptr();
    auto* a = (const unsigned short*)A.data_ptr();
    auto* b = (const unsigned short*)Bt.data_ptr();
    if (N > 64) launch_rr<unsigned short, 128>(c, a, b, rs, M, N, K, relu, s);
    else        launch_rr<unsigned short, 64>(c, a, b, rs, M, N, K, relu, s);
  } else if (A.scalar_type() == torch::kFloat32) {
    auto* c = C.data_ptr<float>();
    auto* a = A.data_ptr<float>();
    auto* b = Bt.data_ptr<float>();
    if (N > 64) launch_rr<float, 128>(c, a, b, rs, M, N, K, relu, s);
    else        launch_rr<float, 64>(c, a, b, rs, M, N, K, relu, s);
  } else {
    TORCH_CHECK(false, "gemm_rr: bf16 or fp32 only");
  }
  ROC_HIP_CHECK(hipGetLastError());
}

void gemm_atb(torch::Tensor C, torch::Tensor A, torch::Tensor B) {
  ROC_CHECK_DEV_CONT(C);
  ROC_CHECK_DEV_CONT(A);
  ROC_CHECK_DEV_CONT(B);
  TORCH_CHECK(A.scalar_type() == B.scalar_type(), "gemm_atb: mixed dtypes");
  TORCH_CHECK(C.scalar_type() == torch::kFloat32, "gemm_atb: fp32 out");
  const int R = (int)A.size(0), Ka = (int)A.size(1);
  const int N = (int)B.size(1);
  TORCH_CHECK(B.size(0) == R, "gemm_atb: row mismatch");
  TORCH_CHECK(C.size(0) == Ka && C.size(1) == N, "gemm_atb: C shape");
  if (R == 0 || Ka == 0 || N == 0) return;  // empty partition: C += 0
  auto s = roc_stream();
  auto* c = C.data_ptr<float>();
  if (A.scalar_type() == torch::kBFloat16) {
    launch_atb<unsigned short>(c, (const unsigned short*)A.data_ptr(),
                               (const unsigned short*)B.data_ptr(), R, Ka, N,
                               s);
  } else if (A.scalar_type() == torch::kFloat32) {
    launch_atb<float>(c, A.data_ptr<float>(), B.data_ptr<float>(), R, Ka, N,
                      s);
  } else {
    TORCH_CHECK(false, "gemm_atb: bf16 or fp32 only");
  }
  ROC_HIP_CHECK(hipGetLastError());
}
