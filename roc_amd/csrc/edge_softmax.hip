// Segment (per-destination-row) softmax over edge scores — the GAT
// attention normalizer. One fused kernel per direction instead of the
// 5-kernel torch composition (scatter-amax, gather, exp, index_add,
// divide): scores are read once, alpha written once.
//
// Geometry: a TEAM of lanes owns one CSR row; lanes stride the row's
// edge segment and fold partials with __shfl_xor trees (teams are
// contiguous power-of-2 lane groups inside a wave64, so the xor tree
// never crosses a team). Three passes in registers/L2: max, exp+sum,
// scale — numerically safe softmax (max-subtracted).
//
// Reference has no edge ops at all (`gnn.cc:475-623` declares edge
// tensors, nothing consumes them); this + spmm_edge make a full
// attention-style aggregation MI355X-native.

#include "common.h"

namespace {

template <int TEAM>
__global__ __launch_bounds__(kBlock) void edge_softmax_fwd_kernel(
    float* __restrict__ alpha, const float* __restrict__ s,
    const int64_t* __restrict__ rowptr, int num_rows) {
  const int tpb = kBlock / TEAM;
  const int team = blockIdx.x * tpb + (int)threadIdx.x / TEAM;
  const int lane = (int)threadIdx.x % TEAM;
  const int nteams = gridDim.x * tpb;
  for (int row = team; row < num_rows; row += nteams) {
    const int64_t e0 = rowptr[row], e1 = rowptr[row + 1];
    if (e0 == e1) continue;
    float m = -3.4e38f;
    for (int64_t e = e0 + lane; e < e1; e += TEAM) m = fmaxf(m, s[e]);
#pragma unroll
    for (int off = TEAM / 2; off > 0; off >>= 1)
      m = fmaxf(m, __shfl_xor(m, off, 64));
    float sum = 0.f;
    for (int64_t e = e0 + lane; e < e1; e += TEAM) {
      const float ex = __expf(s[e] - m);
      alpha[e] = ex;
      sum += ex;
    }
#pragma unroll
    for (int off = TEAM / 2; off > 0; off >>= 1)
      sum += __shfl_xor(sum, off, 64);
    const float inv = 1.f / sum;
    for (int64_t e = e0 + lane; e < e1; e += TEAM) alpha[e] *= inv;
  }
}

// ds[e] = alpha[e] * (dalpha[e] - sum_row(alpha * dalpha))
template <int TEAM>
__global__ __launch_bounds__(kBlock) void edge_softmax_bwd_kernel(
    float* __restrict__ ds, const float* __restrict__ dalpha,
    const float* __restrict__ alpha, const int64_t* __restrict__ rowptr,
    int num_rows) {
  const int tpb = kBlock / TEAM;
  const int team = blockIdx.x * tpb + (int)threadIdx.x / TEAM;
  const int lane = (int)threadIdx.x % TEAM;
  const int nteams = gridDim.x * tpb;
  for (int row = team; row < num_rows; row += nteams) {
    const int64_t e0 = rowptr[row], e1 = rowptr[row + 1];
    if (e0 == e1) continue;
    float dot = 0.f;
    for (int64_t e = e0 + lane; e < e1; e += TEAM)
      dot += alpha[e] * dalpha[e];
#pragma unroll
    for (int off = TEAM / 2; off > 0; off >>= 1)
      dot += __shfl_xor(dot, off, 64);
    for (int64_t e = e0 + lane; e < e1; e += TEAM)
      ds[e] = alpha[e] * (dalpha[e] - dot);
  }
}

}  // namespace

void edge_softmax_fwd(torch::Tensor alpha, torch::Tensor s,
                      torch::Tensor rowptr) {
  ROC_CHECK_DEV_CONT(alpha);
  ROC_CHECK_DEV_CONT(s);
  ROC_CHECK_DEV_CONT(rowptr);
  TORCH_CHECK(s.scalar_type() == torch::kFloat32, "scores must be fp32");
  TORCH_CHECK(alpha.numel() == s.numel(), "size mismatch");
  TORCH_CHECK(rowptr.scalar_type() == torch::kInt64, "rowptr int64");
  const int num_rows = (int)rowptr.numel() - 1;
  constexpr int TEAM = 8;  // avg in-degree ~edges/rows; 8 lanes/row
  const int tpb = kBlock / TEAM;
  hipLaunchKernelGGL((edge_softmax_fwd_kernel<TEAM>),
                     dim3(roc_grid_1d(num_rows, tpb, 8192)), dim3(kBlock),
                     0, roc_stream(), alpha.data_ptr<float>(),
                     s.data_ptr<float>(), rowptr.data_ptr<int64_t>(),
                     num_rows);
  ROC_HIP_CHECK(hipGetLastError());
}

void edge_softmax_bwd(torch::Tensor ds, torch::Tensor dalpha,
                      torch::Tensor alpha, torch::Tensor rowptr) {
  ROC_CHECK_DEV_CONT(ds);
  ROC_CHECK_DEV_CONT(dalpha);
  ROC_CHECK_DEV_CONT(alpha);
  ROC_CHECK_DEV_CONT(rowptr);
  TORCH_CHECK(ds.scalar_type() == torch::kFloat32, "grads must be fp32");
  const int num_rows = (int)rowptr.numel() - 1;
  constexpr int TEAM = 8;
  const int tpb = kBlock / TEAM;
  hipLaunchKernelGGL((edge_softmax_bwd_kernel<TEAM>),
                     dim3(roc_grid_1d(num_rows, tpb, 8192)), dim3(kBlock),
                     0, roc_stream(), ds.data_ptr<float>(),
                     dalpha.data_ptr<float>(), alpha.data_ptr<float>(),
                     rowptr.data_ptr<int64_t>(), num_rows);
  ROC_HIP_CHECK(hipGetLastError());
}

// ---------------------------------------------------------------------------
// Fully-fused GAT attention: score -> LeakyReLU -> segment softmax in
// ONE kernel (and one for the backward), replacing the torch-composed
// chain  s_src[col] + s_dst[row] -> leaky_relu -> edge_softmax
// (4 extra E-length tensors + a 115M-atomic index_add backward).
//   alpha[e] = softmax_row( lrelu(s_src[col_e] + s_dst[row]) )
// Backward emits dsrc (atomicAdd scatter over sources — the split-K dW
// pattern) and dsdst (team-reduced, one store per row).
// ---------------------------------------------------------------------------

namespace {

template <int TEAM>
__global__ __launch_bounds__(kBlock) void att_softmax_fwd_kernel(
    float* __restrict__ alpha, const float* __restrict__ s_src,
    const float* __restrict__ s_dst, const int64_t* __restrict__ rowptr,
    const int* __restrict__ colidx, int num_rows, float slope) {
  const int tpb = kBlock / TEAM;
  const int team = blockIdx.x * tpb + (int)threadIdx.x / TEAM;
  const int lane = (int)threadIdx.x % TEAM;
  const int nteams = gridDim.x * tpb;
  for (int row = team; row < num_rows; row += nteams) {
    const int64_t e0 = rowptr[row], e1 = rowptr[row + 1];
    if (e0 == e1) continue;
    const float sd = s_dst[row];
    float m = -3.4e38f;
    for (int64_t e = e0 + lane; e < e1; e += TEAM) {
      float sc = s_src[colidx[e]] + sd;
      sc = sc > 0.f ? sc : slope * sc;
      alpha[e] = sc;  // stash raw score; normalized below
      m = fmaxf(m, sc);
    }
#pragma unroll
    for (int off = TEAM / 2; off > 0; off >>= 1)
      m = fmaxf(m, __shfl_xor(m, off, 64));
    float sum = 0.f;
    for (int64_t e = e0 + lane; e < e1; e += TEAM) {
      const float ex = __expf(alpha[e] - m);
      alpha[e] = ex;
      sum += ex;
    }
#pragma unroll
    for (int off = TEAM / 2; off > 0; off >>= 1)
      sum += __shfl_xor(sum, off, 64);
    const float inv = 1.f / sum;
    for (int64_t e = e0 + lane; e < e1; e += TEAM) alpha[e] *= inv;
  }
}

// dsrc/dsdst from dalpha: softmax bwd + LeakyReLU grad + scatter.
// dsrc must be zero-initialized (atomic accumulate).
template <int TEAM>
__global__ __launch_bounds__(kBlock) void att_softmax_bwd_kernel(
    float* __restrict__ dsrc, float* __restrict__ dsdst,
    const float* __restrict__ dalpha, const float* __restrict__ alpha,
    const float* __restrict__ s_src, const float* __restrict__ s_dst,
    const int64_t* __restrict__ rowptr, const int* __restrict__ colidx,
    int num_rows, float slope) {
  const int tpb = kBlock / TEAM;
  const int team = blockIdx.x * tpb + (int)threadIdx.x / TEAM;
  const int lane = (int)threadIdx.x % TEAM;
  const int nteams = gridDim.x * tpb;
  for (int row = team; row < num_rows; row += nteams) {
    const int64_t e0 = rowptr[row], e1 = rowptr[row + 1];
    if (e0 == e1) {
      if (lane == 0 && e0 == e1) dsdst[row] = 0.f;
      continue;
    }
    const float sd = s_dst[row];
    float dot = 0.f;
    for (int64_t e = e0 + lane; e < e1; e += TEAM)
      dot += alpha[e] * dalpha[e];
#pragma unroll
    for (int off = TEAM / 2; off > 0; off >>= 1)
      dot += __shfl_xor(dot, off, 64);
    // 4-deep unrolled so the random s_src gathers queue (the rolled
    // version exposed one 4-B gather latency per edge — r2c16)
    float dd = 0.f;
    const int64_t span = e1 - e0;
    const int64_t nfull = span / ((int64_t)TEAM * 4) * ((int64_t)TEAM * 4);
    int64_t e = e0 + lane;
    const int64_t estop = e0 + nfull;
    for (; e < estop; e += (int64_t)TEAM * 4) {
      const int64_t f0 = e, f1 = e + TEAM, f2 = e + 2 * TEAM,
                    f3 = e + 3 * TEAM;
      const int u0 = colidx[f0], u1 = colidx[f1];
      const int u2 = colidx[f2], u3 = colidx[f3];
      const float ss0 = s_src[u0], ss1 = s_src[u1];
      const float ss2 = s_src[u2], ss3 = s_src[u3];
      const float d0 = alpha[f0] * (dalpha[f0] - dot);
      const float d1 = alpha[f1] * (dalpha[f1] - dot);
      const float d2 = alpha[f2] * (dalpha[f2] - dot);
      const float d3 = alpha[f3] * (dalpha[f3] - dot);
      const float g0 = (ss0 + sd) > 0.f ? d0 : slope * d0;
      const float g1 = (ss1 + sd) > 0.f ? d1 : slope * d1;
      const float g2 = (ss2 + sd) > 0.f ? d2 : slope * d2;
      const float g3 = (ss3 + sd) > 0.f ? d3 : slope * d3;
      dd += g0 + g1 + g2 + g3;
      atomicAdd(dsrc + u0, g0);
      atomicAdd(dsrc + u1, g1);
      atomicAdd(dsrc + u2, g2);
      atomicAdd(dsrc + u3, g3);
    }
    for (; e < e1; e += TEAM) {
      const int u = colidx[e];
      const float ds = alpha[e] * (dalpha[e] - dot);
      const float raw = s_src[u] + sd;
      const float dscore = raw > 0.f ? ds : slope * ds;
      dd += dscore;
      atomicAdd(dsrc + u, dscore);
    }
#pragma unroll
    for (int off = TEAM / 2; off > 0; off >>= 1)
      dd += __shfl_xor(dd, off, 64);
    if (lane == 0) dsdst[row] = dd;
  }
}

}  // namespace

void att_softmax_fwd(torch::Tensor alpha, torch::Tensor s_src,
                     torch::Tensor s_dst, torch::Tensor rowptr,
                     torch::Tensor colidx, double slope) {
  ROC_CHECK_DEV_CONT(alpha);
  ROC_CHECK_DEV_CONT(s_src);
  ROC_CHECK_DEV_CONT(s_dst);
  ROC_CHECK_DEV_CONT(rowptr);
  ROC_CHECK_DEV_CONT(colidx);
  TORCH_CHECK(s_src.scalar_type() == torch::kFloat32 &&
              s_dst.scalar_type() == torch::kFloat32, "scores must be fp32");
  TORCH_CHECK(alpha.numel() == colidx.numel(), "size mismatch");
  const int num_rows = (int)rowptr.numel() - 1;
  TORCH_CHECK(s_dst.numel() == num_rows, "s_dst size mismatch");
  constexpr int TEAM = 8;
  const int tpb = kBlock / TEAM;
  hipLaunchKernelGGL((att_softmax_fwd_kernel<TEAM>),
                     dim3(roc_grid_1d(num_rows, tpb, 8192)), dim3(kBlock),
                     0, roc_stream(), alpha.data_ptr<float>(),
                     s_src.data_ptr<float>(), s_dst.data_ptr<float>(),
                     rowptr.data_ptr<int64_t>(), colidx.data_ptr<int>(),
                     num_rows, (float)slope);
  ROC_HIP_CHECK(hipGetLastError());
}

void att_softmax_bwd(torch::Tensor dsrc, torch::Tensor dsdst,
                     torch::Tensor dalpha, torch::Tensor alpha,
                     torch::Tensor s_src, torch::Tensor s_dst,
                     torch::Tensor rowptr, torch::Tensor colidx,
                     double slope) {
  ROC_CHECK_DEV_CONT(dsrc);
  ROC_CHECK_DEV_CONT(dsdst);
  ROC_CHECK_DEV_CONT(dalpha);
  ROC_CHECK_DEV_CONT(alpha);
  const int num_rows = (int)rowptr.numel() - 1;
  TORCH_CHECK(dsrc.numel() == s_src.numel(), "dsrc size mismatch");
  TORCH_CHECK(dsdst.numel() == num_rows, "dsdst size mismatch");
  constexpr int TEAM = 8;
  const int tpb = kBlock / TEAM;
  hipLaunchKernelGGL((att_softmax_bwd_kernel<TEAM>),
                     dim3(roc_grid_1d(num_rows, tpb, 8192)), dim3(kBlock),
                     0, roc_stream(), dsrc.data_ptr<float>(),
                     dsdst.data_ptr<float>(), dalpha.data_ptr<float>(),
                     alpha.data_ptr<float>(), s_src.data_ptr<float>(),
                     s_dst.data_ptr<float>(), rowptr.data_ptr<int64_t>(),
                     colidx.data_ptr<int>(), num_rows, (float)slope);
  ROC_HIP_CHECK(hipGetLastError());
}
