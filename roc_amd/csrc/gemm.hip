// Hand-written MFMA GEMMs for the linear op (CDNA4 / gfx950).
//
//  gemm_rr : C[M,N] = A[M,K] @ B[K,N]        (+ optional fused ReLU)
//            A row-major bf16, B passed PRE-TRANSPOSED as Bt[N,K] so both
//            LDS stages are direct coalesced copies (no transpose writes).
//            Replaces reference cublasSgemm fwd/dX (`linear_kernel.cu:76,227`).
//  gemm_atb: C[Ka,N] += A[R,Ka]^T @ B[R,N]   (fp32 out, split-K atomics)
//            the weight-gradient GEMM (reference `linear_kernel.cu:220`,
//            beta=1 accumulate); reduction dim R is the node count (~10^5-6),
//            so blocks split R and atomically accumulate fp32 partials.
//
// Shapes here are tall-skinny (M ~ nodes, K/N ~ 41..608): the kernels use
// mfma_f32_16x16x32_bf16 with 128-row M-tiles, one K-step of 32, fp32
// accumulation, and zero-filled LDS staging for every edge tail.

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

namespace {

constexpr int BM = 128;   // M rows per block
constexpr int BK = 32;    // K per step (= one MFMA K)
constexpr int PADK = 40;  // LDS row stride in elements (80 B, 16B-aligned)

// Tile staging is split into a LOAD half (global -> regs, issued before
// the MFMAs of the previous tile so HBM latency hides under compute) and
// a WRITE half (regs -> LDS after the barrier). CDNA guide T14.
// 256 threads; thread -> (row, 16-B segment); zero-fill out-of-range.
template <int ROWS, bool ALIGNED>
__device__ __forceinline__ void stage_load(
    short8 (&regs)[ROWS * (BK / 8) / kBlock],
    const unsigned short* __restrict__ g, int row0, int nrows, int64_t ld,
    int k0, int K) {
  constexpr int THREADS_PER_ROW = BK / 8;                  // 4
  constexpr int ROWS_PER_PASS = kBlock / THREADS_PER_ROW;  // 64
  const int seg = threadIdx.x % THREADS_PER_ROW;
  const int r_in = threadIdx.x / THREADS_PER_ROW;
#pragma unroll
  for (int pass = 0; pass < ROWS / ROWS_PER_PASS; ++pass) {
    const int gr = row0 + pass * ROWS_PER_PASS + r_in;
    const int gk = k0 + seg * 8;
    short8 v = {0, 0, 0, 0, 0, 0, 0, 0};
    if (gr < nrows) {
      const unsigned short* p = g + (int64_t)gr * ld + gk;
      if (ALIGNED && gk + 8 <= K) {
        v = *reinterpret_cast<const short8*>(p);
      } else {
        const int nv = min(8, K - gk);
        for (int j = 0; j < nv; ++j) v[j] = (short)p[j];
      }
    }
    regs[pass] = v;
  }
}

template <int ROWS>
__device__ __forceinline__ void stage_write(
    unsigned short* __restrict__ lds,
    const short8 (&regs)[ROWS * (BK / 8) / kBlock]) {
  constexpr int THREADS_PER_ROW = BK / 8;
  constexpr int ROWS_PER_PASS = kBlock / THREADS_PER_ROW;
  const int seg = threadIdx.x % THREADS_PER_ROW;
  const int r_in = threadIdx.x / THREADS_PER_ROW;
#pragma unroll
  for (int pass = 0; pass < ROWS / ROWS_PER_PASS; ++pass) {
    const int r = pass * ROWS_PER_PASS + r_in;
    *reinterpret_cast<short8*>(&lds[r * PADK + seg * 8]) = regs[pass];
  }
}

// ---------------------------------------------------------------------------
// gemm_rr
// ---------------------------------------------------------------------------

template <int BN, bool RELU, bool ALIGNED_A, bool ALIGNED_B>
__global__ __launch_bounds__(kBlock) void gemm_rr_kernel(
    unsigned short* __restrict__ C, const unsigned short* __restrict__ A,
    const unsigned short* __restrict__ Bt, const float* __restrict__ row_scale,
    int M, int N, int K) {
  constexpr int NFRAG = BN / 16;
  __shared__ unsigned short a_lds[2][BM * PADK];
  __shared__ unsigned short b_lds[2][BN * PADK];

  const int m_blk = blockIdx.x * BM;
  const int n_blk = blockIdx.y * BN;
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  const int l15 = lane & 15;
  const int khalf = lane >> 4;  // 0..3 -> k-offset = khalf*8
  const int m_wave = wave * 32;

  f32x4 acc[2][NFRAG];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < NFRAG; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  short8 ra[BM * (BK / 8) / kBlock];
  short8 rb[BN * (BK / 8) / kBlock];
  const int nk = (K + BK - 1) / BK;
  // prologue: tile 0 into LDS buffer 0
  stage_load<BM, ALIGNED_A>(ra, A, m_blk, M, K, 0, K);
  stage_load<BN, ALIGNED_B>(rb, Bt, n_blk, N, K, 0, K);
  stage_write<BM>(a_lds[0], ra);
  stage_write<BN>(b_lds[0], rb);
  __syncthreads();
  int cur = 0;
  for (int kt = 0; kt < nk; ++kt) {
    // issue next tile's global loads before this tile's MFMAs (T14)
    if (kt + 1 < nk) {
      stage_load<BM, ALIGNED_A>(ra, A, m_blk, M, K, (kt + 1) * BK, K);
      stage_load<BN, ALIGNED_B>(rb, Bt, n_blk, N, K, (kt + 1) * BK, K);
    }
#pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
      const int mrow = m_wave + mi * 16 + l15;
      const short8 af = *reinterpret_cast<const short8*>(
          &a_lds[cur][mrow * PADK + khalf * 8]);
#pragma unroll
      for (int ni = 0; ni < NFRAG; ++ni) {
        const int nrow = ni * 16 + l15;
        const short8 bf = *reinterpret_cast<const short8*>(
            &b_lds[cur][nrow * PADK + khalf * 8]);
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af, bf, acc[mi][ni], 0, 0, 0);
      }
    }
    if (kt + 1 < nk) {
      stage_write<BM>(a_lds[cur ^ 1], ra);
      stage_write<BN>(b_lds[cur ^ 1], rb);
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

template <int BN>
void launch_rr(unsigned short* C, const unsigned short* A,
               const unsigned short* Bt, const float* row_scale, int M, int N,
               int K, bool relu, hipStream_t s) {
  dim3 grid((M + BM - 1) / BM, (N + BN - 1) / BN);
  const bool al = (K % 8) == 0;  // both A and Bt have row stride K
#define ROC_RR_CASE(RELU_, ALA)                                              \
  hipLaunchKernelGGL((gemm_rr_kernel<BN, RELU_, ALA, ALA>), grid,            \
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

// stage a [RB x cols] tile TRANSPOSED into LDS[cols][PADK] (lds[c][r]).
template <bool ALIGNED>
__device__ __forceinline__ void stage_tile_T(
    unsigned short* __restrict__ lds, const unsigned short* __restrict__ g,
    int r0, int nrows, int64_t ld, int c0, int ncols) {
  // 256 threads: thread -> (r, cseg) covering RB x 64 elems, 8 per thread
  const int cseg = threadIdx.x % 8;            // 8 elems each
  const int r = threadIdx.x / 8;               // 0..31
  const int gr = r0 + r;
  short8 v = {0, 0, 0, 0, 0, 0, 0, 0};
  if (gr < nrows) {
    const unsigned short* p = g + (int64_t)gr * ld + c0 + cseg * 8;
    const int nv = min(8, ncols - (c0 + cseg * 8));
    if (ALIGNED && nv >= 8) {
      v = *reinterpret_cast<const short8*>(p);
    } else {
      for (int j = 0; j < max(nv, 0); ++j) v[j] = (short)p[j];
    }
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) lds[(cseg * 8 + j) * PADK + r] = (unsigned short)v[j];
}

template <bool ALIGNED_A, bool ALIGNED_B>
__global__ __launch_bounds__(kBlock) void gemm_atb_kernel(
    float* __restrict__ C, const unsigned short* __restrict__ A,
    const unsigned short* __restrict__ B, int R, int Ka, int N,
    int rows_per_split) {
  __shared__ unsigned short at_lds[BKA * PADK];
  __shared__ unsigned short bt_lds[BNW * PADK];

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

  for (int r0 = r_begin; r0 < r_end; r0 += RB) {
    stage_tile_T<ALIGNED_A>(at_lds, A, r0, r_end, Ka, i_blk, Ka);
    stage_tile_T<ALIGNED_B>(bt_lds, B, r0, r_end, N, n_blk, N);
    __syncthreads();
    // zero-fill the r tail inside the step is handled by stage (gr<nrows)
    const int irow = i_wave + l15;
    const short8 af =
        *reinterpret_cast<const short8*>(&at_lds[irow * PADK + khalf * 8]);
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      const short8 bf =
          *reinterpret_cast<const short8*>(&bt_lds[(ni * 16 + l15) * PADK + khalf * 8]);
      acc[ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc[ni], 0, 0, 0);
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

}  // namespace

void gemm_rr(torch::Tensor C, torch::Tensor A, torch::Tensor Bt, bool relu,
             c10::optional<torch::Tensor> row_scale) {
  ROC_CHECK_DEV_CONT(C);
  ROC_CHECK_DEV_CONT(A);
  ROC_CHECK_DEV_CONT(Bt);
  TORCH_CHECK(A.scalar_type() == torch::kBFloat16 &&
                  Bt.scalar_type() == torch::kBFloat16 &&
                  C.scalar_type() == torch::kBFloat16,
              "gemm_rr: bf16 only (GPU compute dtype)");
  const int M = (int)A.size(0), K = (int)A.size(1);
  const int N = (int)Bt.size(0);
  TORCH_CHECK(Bt.size(1) == K, "gemm_rr: Bt must be [N,K]");
  TORCH_CHECK(C.size(0) == M && C.size(1) == N, "gemm_rr: C shape");
  auto s = roc_stream();
  auto* c = (unsigned short*)C.data_ptr();
  auto* a = (const unsigned short*)A.data_ptr();
  auto* b = (const unsigned short*)Bt.data_ptr();
  const float* rs = nullptr;
  if (row_scale.has_value()) {
    TORCH_CHECK(row_scale->scalar_type() == torch::kFloat32 &&
                row_scale->numel() == M, "row_scale must be fp32 [M]");
    rs = row_scale->data_ptr<float>();
  }
  if (N > 64)
    launch_rr<128>(c, a, b, rs, M, N, K, relu, s);
  else
    launch_rr<64>(c, a, b, rs, M, N, K, relu, s);
  ROC_HIP_CHECK(hipGetLastError());
}

void gemm_atb(torch::Tensor C, torch::Tensor A, torch::Tensor B) {
  ROC_CHECK_DEV_CONT(C);
  ROC_CHECK_DEV_CONT(A);
  ROC_CHECK_DEV_CONT(B);
  TORCH_CHECK(A.scalar_type() == torch::kBFloat16 &&
                  B.scalar_type() == torch::kBFloat16,
              "gemm_atb: bf16 inputs");
  TORCH_CHECK(C.scalar_type() == torch::kFloat32, "gemm_atb: fp32 out");
  const int R = (int)A.size(0), Ka = (int)A.size(1);
  const int N = (int)B.size(1);
  TORCH_CHECK(B.size(0) == R, "gemm_atb: row mismatch");
  TORCH_CHECK(C.size(0) == Ka && C.size(1) == N, "gemm_atb: C shape");
  const int tiles = ((Ka + BKA - 1) / BKA) * ((N + BNW - 1) / BNW);
  int splitk = 2048 / max(tiles, 1);
  splitk = max(1, min(splitk, (R + RB - 1) / RB));
  const char* det = getenv("ROC_DETERMINISTIC");
  if (det && det[0] == '1') splitk = 1;  // bit-reproducible dW (slower)
  int rows_per_split = ((R + splitk - 1) / splitk + RB - 1) / RB * RB;
  splitk = (R + rows_per_split - 1) / rows_per_split;
  dim3 grid((Ka + BKA - 1) / BKA, (N + BNW - 1) / BNW, splitk);
  const bool ala = (Ka % 8) == 0;
  const bool alb = (N % 8) == 0;
  auto s = roc_stream();
  auto* c = C.data_ptr<float>();
  auto* a = (const unsigned short*)A.data_ptr();
  auto* b = (const unsigned short*)B.data_ptr();
#define ROC_ATB_CASE(ALA, ALB)                                              \
  hipLaunchKernelGGL((gemm_atb_kernel<ALA, ALB>), grid, dim3(kBlock), 0, s, \
                     c, a, b, R, Ka, N, rows_per_split)
  if (ala) { if (alb) ROC_ATB_CASE(true, true); else ROC_ATB_CASE(true, false); }
  else     { if (alb) ROC_ATB_CASE(false, true); else ROC_ATB_CASE(false, false); }
#undef ROC_ATB_CASE
  ROC_HIP_CHECK(hipGetLastError());
}
