// CSR SpMM neighbor aggregation for CDNA4 — the hot GNN op.
//
// out[v, :] = (deg_dst[v] *) sum_{e in row v} (deg_src[u_e] *) x[u_e, :]
//
// MI355X-native design (vs reference `scattergather_kernel.cu:20-76`,
// which staged the full graph tensor host->FB per task and used
// cub-scan + LDS atomics):
//  - features stay HBM/L3-resident; no host staging
//  - a "row team" of TEAM lanes owns one row at a time; each lane
//    accumulates EPU output elements in registers (no LDS, no atomics)
//  - 16-B vectorized gathers (uint4 of 8 bf16 / float4), 4-deep unrolled
//  - optional fused degree scaling (deg_* are precomputed factors); the
//    fast GCN path pre-scales sources in the GEMM epilogue so only the
//    cheap per-row dst factor remains here
//  - rows are visited in degree-descending order (row_order) so hub rows
//    start first and the skew tail is hidden (Reddit max degree >> mean)
//  - wide feature dims are column-tiled across blockIdx.y
//
// The vector path and the (rare) partial-unit tail path are fully
// separated: the hot path's per-slot buffers are only ever statically
// indexed, so nothing lands in scratch (a shared runtime-indexed array
// cost 144 B/lane of scratch and 3x wall time — see
// profiles/r02 notes / CDNA guide §5.4 rule 20).
// Backward runs this same kernel on the transpose CSR (exact on
// asymmetric graphs; the reference assumed symmetry).

#include "common.h"

namespace {

// raw 16-B gather payload: unpacking is deferred to accumulate time so
// an in-flight slot costs 4 VGPRs, not 8 (occupancy = latency hiding)
template <typename T>
struct RawVec { using type = float4; };
template <>
struct RawVec<unsigned short> { using type = uint4; };

__device__ __forceinline__ uint4 load_raw(const unsigned short* p) {
  return *reinterpret_cast<const uint4*>(p);
}
__device__ __forceinline__ float4 load_raw(const float* p) {
  return *reinterpret_cast<const float4*>(p);
}
__device__ __forceinline__ void acc_add(float* acc, const uint4& r, float w) {
  acc[0] += w * bf16_lo(r.x); acc[1] += w * bf16_hi(r.x);
  acc[2] += w * bf16_lo(r.y); acc[3] += w * bf16_hi(r.y);
  acc[4] += w * bf16_lo(r.z); acc[5] += w * bf16_hi(r.z);
  acc[6] += w * bf16_lo(r.w); acc[7] += w * bf16_hi(r.w);
}
__device__ __forceinline__ void acc_add(float* acc, const float4& r, float w) {
  acc[0] += w * r.x; acc[1] += w * r.y; acc[2] += w * r.z; acc[3] += w * r.w;
}

typedef __attribute__((ext_vector_type(4))) float f32x4v;

// Two gather engines: plain global loads (64-bit addressing), and SRSRC
// buffer loads with 32-bit offsets when the matrix fits 4 GB — fewer
// address VGPRs = more waves in flight on a latency-bound kernel
// (CDNA guide T8/T20; the descriptor is built from kernargs only, so no
// waterfall loops).
template <typename T>
struct GlobalGather {
  const T* x;
  int64_t D;
  int64_t col0;
  __device__ __forceinline__ typename RawVec<T>::type load(int u) const {
    return load_raw(x + (int64_t)u * D + col0);
  }
};

template <typename T>
struct BufferGather {
  __amdgpu_buffer_rsrc_t rsrc;
  unsigned dbytes;    // D * sizeof(T)
  unsigned colbytes;  // col0 * sizeof(T)
  __device__ __forceinline__ typename RawVec<T>::type load(int u) const {
    const f32x4v v = __builtin_amdgcn_raw_buffer_load_b128(
        rsrc, (unsigned)u * dbytes + colbytes, 0, 0);
    return __builtin_bit_cast(typename RawVec<T>::type, v);
  }
};

// hot path: this lane's 16-B unit is entirely inside D.
// SRC is compile-time: the fused GCN path has no per-edge source scale,
// and dropping the dead deg_src lookups frees the 8 index registers
// (they otherwise stay live across all 8 in-flight gathers).
template <typename T, int EPU, int UN, bool SRC, typename LD>
__device__ __forceinline__ void row_accum_vec(
    float* __restrict__ acc, const LD& ld,
    const int* __restrict__ colidx, const float* __restrict__ deg_src,
    int64_t e0, int64_t e1) {
  using Raw = typename RawVec<T>::type;
  int64_t e = e0;
  if (UN >= 16) {
    // 16 in flight: two 8-groups issued back-to-back
    for (; e + 15 < e1; e += 16) {
      const int u0 = colidx[e], u1 = colidx[e + 1];
      const int u2 = colidx[e + 2], u3 = colidx[e + 3];
      const int u4 = colidx[e + 4], u5 = colidx[e + 5];
      const int u6 = colidx[e + 6], u7 = colidx[e + 7];
      const int v0 = colidx[e + 8], v1 = colidx[e + 9];
      const int v2 = colidx[e + 10], v3 = colidx[e + 11];
      const int v4 = colidx[e + 12], v5 = colidx[e + 13];
      const int v6 = colidx[e + 14], v7 = colidx[e + 15];
      const Raw r0 = ld.load(u0), r1 = ld.load(u1);
      const Raw r2 = ld.load(u2), r3 = ld.load(u3);
      const Raw r4 = ld.load(u4), r5 = ld.load(u5);
      const Raw r6 = ld.load(u6), r7 = ld.load(u7);
      const Raw s0 = ld.load(v0), s1 = ld.load(v1);
      const Raw s2 = ld.load(v2), s3 = ld.load(v3);
      const Raw s4 = ld.load(v4), s5 = ld.load(v5);
      const Raw s6 = ld.load(v6), s7 = ld.load(v7);
      if (SRC) {
        acc_add(acc, r0, deg_src[u0]); acc_add(acc, r1, deg_src[u1]);
        acc_add(acc, r2, deg_src[u2]); acc_add(acc, r3, deg_src[u3]);
        acc_add(acc, r4, deg_src[u4]); acc_add(acc, r5, deg_src[u5]);
        acc_add(acc, r6, deg_src[u6]); acc_add(acc, r7, deg_src[u7]);
        acc_add(acc, s0, deg_src[v0]); acc_add(acc, s1, deg_src[v1]);
        acc_add(acc, s2, deg_src[v2]); acc_add(acc, s3, deg_src[v3]);
        acc_add(acc, s4, deg_src[v4]); acc_add(acc, s5, deg_src[v5]);
        acc_add(acc, s6, deg_src[v6]); acc_add(acc, s7, deg_src[v7]);
      } else {
        acc_add(acc, r0, 1.f); acc_add(acc, r1, 1.f);
        acc_add(acc, r2, 1.f); acc_add(acc, r3, 1.f);
        acc_add(acc, r4, 1.f); acc_add(acc, r5, 1.f);
        acc_add(acc, r6, 1.f); acc_add(acc, r7, 1.f);
        acc_add(acc, s0, 1.f); acc_add(acc, s1, 1.f);
        acc_add(acc, s2, 1.f); acc_add(acc, s3, 1.f);
        acc_add(acc, s4, 1.f); acc_add(acc, s5, 1.f);
        acc_add(acc, s6, 1.f); acc_add(acc, s7, 1.f);
      }
    }
  }
  if (UN >= 8) {
    // 8 in flight (raw payloads, unpack at use)
    for (; e + 7 < e1; e += 8) {
      const int u0 = colidx[e], u1 = colidx[e + 1];
      const int u2 = colidx[e + 2], u3 = colidx[e + 3];
      const int u4 = colidx[e + 4], u5 = colidx[e + 5];
      const int u6 = colidx[e + 6], u7 = colidx[e + 7];
      const Raw r0 = ld.load(u0);
      const Raw r1 = ld.load(u1);
      const Raw r2 = ld.load(u2);
      const Raw r3 = ld.load(u3);
      const Raw r4 = ld.load(u4);
      const Raw r5 = ld.load(u5);
      const Raw r6 = ld.load(u6);
      const Raw r7 = ld.load(u7);
      if (SRC) {
        acc_add(acc, r0, deg_src[u0]); acc_add(acc, r1, deg_src[u1]);
        acc_add(acc, r2, deg_src[u2]); acc_add(acc, r3, deg_src[u3]);
        acc_add(acc, r4, deg_src[u4]); acc_add(acc, r5, deg_src[u5]);
        acc_add(acc, r6, deg_src[u6]); acc_add(acc, r7, deg_src[u7]);
      } else {
        acc_add(acc, r0, 1.f); acc_add(acc, r1, 1.f);
        acc_add(acc, r2, 1.f); acc_add(acc, r3, 1.f);
        acc_add(acc, r4, 1.f); acc_add(acc, r5, 1.f);
        acc_add(acc, r6, 1.f); acc_add(acc, r7, 1.f);
      }
    }
  }
  for (; e + 3 < e1; e += 4) {
    const int u0 = colidx[e], u1 = colidx[e + 1];
    const int u2 = colidx[e + 2], u3 = colidx[e + 3];
    const Raw r0 = ld.load(u0);
    const Raw r1 = ld.load(u1);
    const Raw r2 = ld.load(u2);
    const Raw r3 = ld.load(u3);
    if (SRC) {
      acc_add(acc, r0, deg_src[u0]); acc_add(acc, r1, deg_src[u1]);
      acc_add(acc, r2, deg_src[u2]); acc_add(acc, r3, deg_src[u3]);
    } else {
      acc_add(acc, r0, 1.f); acc_add(acc, r1, 1.f);
      acc_add(acc, r2, 1.f); acc_add(acc, r3, 1.f);
    }
  }
  for (; e < e1; ++e) {
    const int u0 = colidx[e];
    const Raw r0 = ld.load(u0);
    acc_add(acc, r0, SRC ? deg_src[u0] : 1.f);
  }
}

// tail path: unit straddles the column window (only when its width is
// not a multiple of EPU; engine pads dims to 8 so this is cold).
// Scalar loads, dynamic-bound loops.
template <typename T, int EPU>
__device__ void row_accum_tail(
    float* __restrict__ acc, const T* __restrict__ x, int64_t ld, int64_t col0,
    int nvalid, const int* __restrict__ colidx,
    const float* __restrict__ deg_src, int64_t e0, int64_t e1) {
  for (int64_t e = e0; e < e1; ++e) {
    const int u0 = colidx[e];
    const T* r = x + (int64_t)u0 * ld + col0;
    const float w0 = deg_src ? deg_src[u0] : 1.f;
    for (int j = 0; j < nvalid; ++j) acc[j] += w0 * elt_to_f32(r[j]);
  }
}

// (a forced 6-waves/SIMD __launch_bounds__ hint was measured NEUTRAL:
//  the allocator spills 28 B/lane on half the hot variants and any
//  occupancy gain washes out — keep the default allocation)
// OT (output type) may differ from T: OT=float with T=bf16 is the
// strip-blocked accumulation path — fp32 partials in HBM, one rounding
// at the final cast instead of one per strip pass.
// ld is the row stride; [col_base, col_end) is the column window this
// launch covers (ld == col_end, col_base == 0 for a whole-matrix pass;
// a column-phase pass sweeps a 64-col window of a wider matrix so the
// fp32 partial buffer is touched once per much-wider source strip).
template <typename T, typename OT, int TEAM, int UN, bool BUF, bool SRC>
__global__ __launch_bounds__(kBlock) void spmm_kernel(
    OT* __restrict__ out, const T* __restrict__ x,
    const int64_t* __restrict__ rowptr, const int* __restrict__ colidx,
    const float* __restrict__ deg_dst, const float* __restrict__ deg_src,
    const int* __restrict__ row_order, int num_rows, int64_t ld,
    int64_t col_base, int64_t col_end, bool accumulate, unsigned x_bytes) {
  constexpr int EPU = EltTraits<T>::kPerVec;
  const int tpb = kBlock / TEAM;
  const int team = blockIdx.x * tpb + (int)threadIdx.x / TEAM;
  const int lane = (int)threadIdx.x % TEAM;
  const int nteams = gridDim.x * tpb;

  const int64_t col0 =
      col_base + ((int64_t)blockIdx.y * TEAM + lane) * EPU;
  const bool full = (col0 + EPU) <= col_end;
  const int nvalid =
      full ? EPU : (col0 < col_end ? (int)(col_end - col0) : 0);

  if (full) {
    for (int ri = team; ri < num_rows; ri += nteams) {
      const int row = row_order ? row_order[ri] : ri;
      const int64_t e0 = rowptr[row];
      const int64_t e1 = rowptr[row + 1];
      float acc[EPU];
      OT* o = out + (int64_t)row * ld + col0;
      if (accumulate) {  // strip / halo-overlap pass 2: from partials
#pragma unroll
        for (int j = 0; j < EPU; ++j) acc[j] = elt_to_f32(o[j]);
      } else {
#pragma unroll
        for (int j = 0; j < EPU; ++j) acc[j] = 0.f;
      }
      if constexpr (BUF) {
        BufferGather<T> g{
            __builtin_amdgcn_make_buffer_rsrc((void*)x, (short)0, x_bytes,
                                              0x00020000),
            (unsigned)(ld * sizeof(T)), (unsigned)(col0 * sizeof(T))};
        row_accum_vec<T, EPU, UN, SRC>(acc, g, colidx, deg_src, e0, e1);
      } else {
        GlobalGather<T> g{x, ld, col0};
        row_accum_vec<T, EPU, UN, SRC>(acc, g, colidx, deg_src, e0, e1);
      }
      if (deg_dst) {
        const float s = deg_dst[row];
#pragma unroll
        for (int j = 0; j < EPU; ++j) acc[j] *= s;
      }
      if constexpr (std::is_same<OT, unsigned short>::value) {
        store_bf16x8(o, acc);
      } else if constexpr (EPU == 8) {
        store_f32x8(o, acc);
      } else {
        store_f32x4(o, acc);
      }
    }
  } else if (nvalid > 0) {
    for (int ri = team; ri < num_rows; ri += nteams) {
      const int row = row_order ? row_order[ri] : ri;
      float acc[EPU];
      OT* o = out + (int64_t)row * ld + col0;
      for (int j = 0; j < nvalid; ++j)
        acc[j] = accumulate ? elt_to_f32(o[j]) : 0.f;
      row_accum_tail<T, EPU>(acc, x, ld, col0, nvalid, colidx, deg_src,
                             rowptr[row], rowptr[row + 1]);
      const float s = deg_dst ? deg_dst[row] : 1.f;
      for (int j = 0; j < nvalid; ++j) f32_to_elt(acc[j] * s, o + j);
    }
  }
}

// A/B geometry knobs, pinned ONCE at first launch (a per-launch getenv
// on the hot path is a footgun under hipGraph capture: replays would
// silently keep whatever value was live at capture; now ALL launches —
// captured or eager — use the same process-lifetime values).
struct SpmmKnobs {
  int team_override;  // 0 = auto (by D); else 8/16/32/64
  int unroll;         // 4 | 8 | 16 (16 measured best)
  bool allow_buffer;  // SRSRC buffer_load gather path
};

static SpmmKnobs read_spmm_knobs() {
  SpmmKnobs k{0, 16, true};
  if (const char* e = getenv("ROC_SPMM_TEAM")) {
    const int t = atoi(e);
    if (t == 8 || t == 16 || t == 32 || t == 64) k.team_override = t;
  }
  if (const char* e = getenv("ROC_SPMM_UNROLL")) k.unroll = atoi(e);
  if (const char* e = getenv("ROC_SPMM_BUFFER")) k.allow_buffer = e[0] != '0';
  return k;
}

static SpmmKnobs g_spmm_knobs;
static bool g_spmm_knobs_read = false;

static const SpmmKnobs& spmm_knobs() {
  if (!g_spmm_knobs_read) {
    g_spmm_knobs = read_spmm_knobs();
    g_spmm_knobs_read = true;
  }
  return g_spmm_knobs;
}

template <typename T, typename OT>
void launch_spmm(OT* out, const T* x, const int64_t* rowptr,
                 const int* colidx, const float* deg_dst,
                 const float* deg_src, const int* row_order, int num_rows,
                 int64_t ld, int64_t col_base, int64_t col_end,
                 bool accumulate, size_t x_elems, hipStream_t stream) {
  constexpr int EPU = EltTraits<T>::kPerVec;
  const SpmmKnobs& kn = spmm_knobs();
  const int64_t ncols = col_end - col_base;
  const int64_t units = (ncols + EPU - 1) / EPU;
  int team = 8;
  while (team < units && team < 64) team *= 2;
  if (kn.team_override) team = kn.team_override;
  const int col_tiles = (int)((units + team - 1) / team);
  const int tpb = kBlock / team;
  dim3 grid(roc_grid_1d(num_rows, tpb, 8192), col_tiles);
  const int un = kn.unroll;
  const size_t xb = x_elems * sizeof(T);
  const bool buf = xb < (size_t)UINT_MAX && kn.allow_buffer;
  const unsigned x_bytes = (unsigned)(buf ? xb : 0);
  // measured (scripts/bench_spmm.py, Reddit shape): degree-descending
  // scheduling wins for wide rows (D=256: -13%) but loses for narrow
  // ones (D=48: +10% — the indirection costs more than the skew tail)
  if (team < 16) row_order = nullptr;
#define ROC_SPMM_L3(TEAM_, UN_, BUF_)                                       \
  do {                                                                      \
    if (deg_src) {                                                          \
      hipLaunchKernelGGL((spmm_kernel<T, OT, TEAM_, UN_, BUF_, true>),     \
                         grid,                                              \
                         dim3(kBlock), 0, stream, out, x, rowptr, colidx,   \
                         deg_dst, deg_src, row_order, num_rows, ld,         \
                         col_base, col_end, accumulate, x_bytes);           \
    } else {                                                                \
      hipLaunchKernelGGL((spmm_kernel<T, OT, TEAM_, UN_, BUF_, false>),    \
                         grid,                                              \
                         dim3(kBlock), 0, stream, out, x, rowptr, colidx,   \
                         deg_dst, deg_src, row_order, num_rows, ld,         \
                         col_base, col_end, accumulate, x_bytes);           \
    }                                                                       \
  } while (0)
#define ROC_SPMM_L2(TEAM_)                                                  \
  do {                                                                      \
    if (buf) {                                                              \
      if (un >= 16) { ROC_SPMM_L3(TEAM_, 16, true); }                       \
      else if (un >= 8) { ROC_SPMM_L3(TEAM_, 8, true); }                    \
      else { ROC_SPMM_L3(TEAM_, 4, true); }                                 \
    } else {                                                                \
      if (un >= 16) { ROC_SPMM_L3(TEAM_, 16, false); }                      \
      else if (un >= 8) { ROC_SPMM_L3(TEAM_, 8, false); }                   \
      else { ROC_SPMM_L3(TEAM_, 4, false); }                                \
    }                                                                       \
  } while (0)
  switch (team) {
    case 8:  ROC_SPMM_L2(8);  break;
    case 16: ROC_SPMM_L2(16); break;
    case 32: ROC_SPMM_L2(32); break;
    default: ROC_SPMM_L2(64);
  }
#undef ROC_SPMM_L2
#undef ROC_SPMM_L3
}

// ---------------------------------------------------------------------------
// Edge-weighted SpMM + its edge-value gradient (the edge-tensor ops).
// out[v,:] (+)= deg_dst[v] * sum_{e in row v} edge_val[e] * x[col_e,:]
// Same row-team geometry as spmm_kernel; weights are per-EDGE (not
// per-source like deg_src), loaded sequentially -> one scalar broadcast
// per gather. 4-deep unrolled; capability path, not the fused-GCN hot
// path, so one unroll depth is enough.
// ---------------------------------------------------------------------------

// per-row accumulate loop shared by both gather engines (8 gathers in
// flight via 2x 4-group issue; per-edge scalar weight broadcast)
template <typename T, int EPU, typename LD>
__device__ __forceinline__ void edge_accum_row(
    float* __restrict__ acc, const LD& ld, const int* __restrict__ colidx,
    const float* __restrict__ edge_val, int64_t e, int64_t e1) {
  using Raw = typename RawVec<T>::type;
  for (; e + 3 < e1; e += 4) {
    const int u0 = colidx[e], u1 = colidx[e + 1];
    const int u2 = colidx[e + 2], u3 = colidx[e + 3];
    const float w0 = edge_val[e], w1 = edge_val[e + 1];
    const float w2 = edge_val[e + 2], w3 = edge_val[e + 3];
    const Raw r0 = ld.load(u0), r1 = ld.load(u1);
    const Raw r2 = ld.load(u2), r3 = ld.load(u3);
    acc_add(acc, r0, w0); acc_add(acc, r1, w1);
    acc_add(acc, r2, w2); acc_add(acc, r3, w3);
  }
  for (; e < e1; ++e) {
    const Raw r0 = ld.load(colidx[e]);
    acc_add(acc, r0, edge_val[e]);
  }
}

template <typename T, int TEAM, bool BUF>
__global__ __launch_bounds__(kBlock) void spmm_edge_kernel(
    T* __restrict__ out, const T* __restrict__ x,
    const int64_t* __restrict__ rowptr, const int* __restrict__ colidx,
    const float* __restrict__ edge_val, const float* __restrict__ deg_dst,
    const int* __restrict__ row_order, int num_rows, int64_t D,
    bool accumulate, unsigned x_bytes) {
  constexpr int EPU = EltTraits<T>::kPerVec;
  const int tpb = kBlock / TEAM;
  const int team = blockIdx.x * tpb + (int)threadIdx.x / TEAM;
  const int lane = (int)threadIdx.x % TEAM;
  const int nteams = gridDim.x * tpb;
  const int64_t col0 = ((int64_t)blockIdx.y * TEAM + lane) * EPU;
  const bool full = (col0 + EPU) <= D;
  const int nvalid = full ? EPU : (col0 < D ? (int)(D - col0) : 0);

  if (full) {
    for (int ri = team; ri < num_rows; ri += nteams) {
      const int row = row_order ? row_order[ri] : ri;
      const int64_t e0 = rowptr[row];
      const int64_t e1 = rowptr[row + 1];
      float acc[EPU];
      T* o = out + (int64_t)row * D + col0;
      if (accumulate) {
#pragma unroll
        for (int j = 0; j < EPU; ++j) acc[j] = elt_to_f32(o[j]);
      } else {
#pragma unroll
        for (int j = 0; j < EPU; ++j) acc[j] = 0.f;
      }
      if constexpr (BUF) {
        BufferGather<T> ld{
            __builtin_amdgcn_make_buffer_rsrc((void*)x, (short)0, x_bytes,
                                              0x00020000),
            (unsigned)(D * sizeof(T)), (unsigned)(col0 * sizeof(T))};
        edge_accum_row<T, EPU>(acc, ld, colidx, edge_val, e0, e1);
      } else {
        GlobalGather<T> ld{x, D, col0};
        edge_accum_row<T, EPU>(acc, ld, colidx, edge_val, e0, e1);
      }
      if (deg_dst) {
        const float s = deg_dst[row];
#pragma unroll
        for (int j = 0; j < EPU; ++j) acc[j] *= s;
      }
      if constexpr (EPU == 8) store_bf16x8(o, acc); else store_f32x4(o, acc);
    }
  } else if (nvalid > 0) {
    for (int ri = team; ri < num_rows; ri += nteams) {
      const int row = row_order ? row_order[ri] : ri;
      float acc[EPU];
      T* o = out + (int64_t)row * D + col0;
      for (int j = 0; j < nvalid; ++j)
        acc[j] = accumulate ? elt_to_f32(o[j]) : 0.f;
      for (int64_t e = rowptr[row]; e < rowptr[row + 1]; ++e) {
        const T* r = x + (int64_t)colidx[e] * D + col0;
        const float w0 = edge_val[e];
        for (int j = 0; j < nvalid; ++j) acc[j] += w0 * elt_to_f32(r[j]);
      }
      const float s = deg_dst ? deg_dst[row] : 1.f;
      for (int j = 0; j < nvalid; ++j) f32_to_elt(acc[j] * s, o + j);
    }
  }
}

// dw[e] = <dy[row_e,:], x[col_e,:]>  (fp32 accumulate).
// Team per row; per edge each lane dots its strided D-chunks, then a
// log2(TEAM) shfl_xor tree folds the team partials; lane 0 stores.
// Teams are contiguous, power-of-2-aligned lane groups, so the xor
// tree never crosses a team boundary within the wave64 front.
template <typename T, int TEAM>
__global__ __launch_bounds__(kBlock) void edge_dot_kernel(
    float* __restrict__ dw, const T* __restrict__ dy,
    const T* __restrict__ x, const int64_t* __restrict__ rowptr,
    const int* __restrict__ colidx, int num_rows, int64_t D) {
  constexpr int EPU = EltTraits<T>::kPerVec;
  using Raw = typename RawVec<T>::type;
  const int tpb = kBlock / TEAM;
  const int team = blockIdx.x * tpb + (int)threadIdx.x / TEAM;
  const int lane = (int)threadIdx.x % TEAM;
  const int nteams = gridDim.x * tpb;
  // fast path: one chunk covers D (TEAM*EPU >= D, e.g. TEAM=8 D<=64
  // bf16). The dy row chunk is loaded ONCE per row (hoisted out of the
  // edge loop) and x gathers are queued 4 deep — the naive per-edge
  // load/reduce version was 4x off the gather-service rate (r2c16).
  if ((int64_t)TEAM * EPU >= D && D % EPU == 0) {
    const int cvalid = (int)(D / EPU);  // lanes with a live chunk
    for (int row = team; row < num_rows; row += nteams) {
      const int64_t e0 = rowptr[row], e1 = rowptr[row + 1];
      const T* dyr = dy + (int64_t)row * D;
      float a[EPU] = {0.f};
      if (lane < cvalid) {
        const Raw ra = load_raw(dyr + (int64_t)lane * EPU);
        acc_add(a, ra, 1.f);
      }
      int64_t e = e0;
      for (; e + 3 < e1; e += 4) {
        const int u0 = colidx[e], u1 = colidx[e + 1];
        const int u2 = colidx[e + 2], u3 = colidx[e + 3];
        float p0 = 0.f, p1 = 0.f, p2 = 0.f, p3 = 0.f;
        if (lane < cvalid) {
          const int64_t c = (int64_t)lane * EPU;
          const Raw r0 = load_raw(x + (int64_t)u0 * D + c);
          const Raw r1 = load_raw(x + (int64_t)u1 * D + c);
          const Raw r2 = load_raw(x + (int64_t)u2 * D + c);
          const Raw r3 = load_raw(x + (int64_t)u3 * D + c);
          float b[EPU];
#pragma unroll
          for (int j = 0; j < EPU; ++j) b[j] = 0.f;
          acc_add(b, r0, 1.f);
#pragma unroll
          for (int j = 0; j < EPU; ++j) p0 += a[j] * b[j];
#pragma unroll
          for (int j = 0; j < EPU; ++j) b[j] = 0.f;
          acc_add(b, r1, 1.f);
#pragma unroll
          for (int j = 0; j < EPU; ++j) p1 += a[j] * b[j];
#pragma unroll
          for (int j = 0; j < EPU; ++j) b[j] = 0.f;
          acc_add(b, r2, 1.f);
#pragma unroll
          for (int j = 0; j < EPU; ++j) p2 += a[j] * b[j];
#pragma unroll
          for (int j = 0; j < EPU; ++j) b[j] = 0.f;
          acc_add(b, r3, 1.f);
#pragma unroll
          for (int j = 0; j < EPU; ++j) p3 += a[j] * b[j];
        }
#pragma unroll
        for (int off = TEAM / 2; off > 0; off >>= 1) {
          p0 += __shfl_xor(p0, off, 64);
          p1 += __shfl_xor(p1, off, 64);
          p2 += __shfl_xor(p2, off, 64);
          p3 += __shfl_xor(p3, off, 64);
        }
        if (lane == 0) {
          dw[e] = p0; dw[e + 1] = p1; dw[e + 2] = p2; dw[e + 3] = p3;
        }
      }
      for (; e < e1; ++e) {
        float p = 0.f;
        if (lane < cvalid) {
          const Raw rb = load_raw(x + (int64_t)colidx[e] * D
                                  + (int64_t)lane * EPU);
          float b[EPU];
#pragma unroll
          for (int j = 0; j < EPU; ++j) b[j] = 0.f;
          acc_add(b, rb, 1.f);
#pragma unroll
          for (int j = 0; j < EPU; ++j) p += a[j] * b[j];
        }
#pragma unroll
        for (int off = TEAM / 2; off > 0; off >>= 1)
          p += __shfl_xor(p, off, 64);
        if (lane == 0) dw[e] = p;
      }
    }
    return;
  }
  // general path (wide D): chunk loop per edge
  for (int row = team; row < num_rows; row += nteams) {
    const int64_t e0 = rowptr[row], e1 = rowptr[row + 1];
    const T* dyr = dy + (int64_t)row * D;
    for (int64_t e = e0; e < e1; ++e) {
      const T* xr = x + (int64_t)colidx[e] * D;
      float p = 0.f;
      for (int64_t c = (int64_t)lane * EPU; c + EPU <= D;
           c += (int64_t)TEAM * EPU) {
        const Raw a = load_raw(dyr + c);
        const Raw b = load_raw(xr + c);
        float ta[EPU] = {0.f}, tb[EPU] = {0.f};
        acc_add(ta, a, 1.f);
        acc_add(tb, b, 1.f);
#pragma unroll
        for (int j = 0; j < EPU; ++j) p += ta[j] * tb[j];
      }
      // scalar tail, lane-strided, for D % EPU != 0 (cold; dims padded)
      for (int64_t j = (D / EPU) * EPU + lane; j < D; j += TEAM)
        p += elt_to_f32(dyr[j]) * elt_to_f32(xr[j]);
#pragma unroll
      for (int off = TEAM / 2; off > 0; off >>= 1)
        p += __shfl_xor(p, off, 64);
      if (lane == 0) dw[e] = p;
    }
  }
}

template <typename T>
void launch_spmm_edge(T* out, const T* x, const int64_t* rowptr,
                      const int* colidx, const float* edge_val,
                      const float* deg_dst, const int* row_order,
                      int num_rows, int64_t D, bool accumulate,
                      size_t x_elems, hipStream_t stream) {
  constexpr int EPU = EltTraits<T>::kPerVec;
  const int64_t units = (D + EPU - 1) / EPU;
  int team = 8;
  while (team < units && team < 64) team *= 2;
  const int col_tiles = (int)((units + team - 1) / team);
  const int tpb = kBlock / team;
  dim3 grid(roc_grid_1d(num_rows, tpb, 8192), col_tiles);
  const size_t xb = x_elems * sizeof(T);
  const bool buf = xb < (size_t)UINT_MAX && spmm_knobs().allow_buffer;
  if (team < 16) row_order = nullptr;
#define ROC_SPMME_L(TEAM_, BUF_)                                            \
  hipLaunchKernelGGL((spmm_edge_kernel<T, TEAM_, BUF_>), grid, dim3(kBlock),\
                     0, stream, out, x, rowptr, colidx, edge_val, deg_dst,  \
                     row_order, num_rows, D, accumulate,                    \
                     (unsigned)(buf ? xb : 0))
#define ROC_SPMME_T(TEAM_)                                                  \
  do {                                                                      \
    if (buf) { ROC_SPMME_L(TEAM_, true); }                                  \
    else { ROC_SPMME_L(TEAM_, false); }                                     \
  } while (0)
  switch (team) {
    case 8:  ROC_SPMME_T(8);  break;
    case 16: ROC_SPMME_T(16); break;
    case 32: ROC_SPMME_T(32); break;
    default: ROC_SPMME_T(64);
  }
#undef ROC_SPMME_T
#undef ROC_SPMME_L
}

template <typename T>
void launch_edge_dot(float* dw, const T* dy, const T* x,
                     const int64_t* rowptr, const int* colidx, int num_rows,
                     int64_t D, hipStream_t stream) {
  constexpr int EPU = EltTraits<T>::kPerVec;
  const int64_t units = (D + EPU - 1) / EPU;
  int team = 8;
  while (team < units && team < 64) team *= 2;
  const int tpb = kBlock / team;
  dim3 grid(roc_grid_1d(num_rows, tpb, 8192));
#define ROC_EDOT_L(TEAM_)                                                   \
  hipLaunchKernelGGL((edge_dot_kernel<T, TEAM_>), grid, dim3(kBlock), 0,    \
                     stream, dw, dy, x, rowptr, colidx, num_rows, D)
  switch (team) {
    case 8:  ROC_EDOT_L(8);  break;
    case 16: ROC_EDOT_L(16); break;
    case 32: ROC_EDOT_L(32); break;
    default: ROC_EDOT_L(64);
  }
#undef ROC_EDOT_L
}

}  // namespace

// A/B harness hook (scripts/bench_spmm.py): re-read the geometry env
// vars. Never call while a hipGraph capture of spmm launches is live.
void spmm_refresh_knobs() { g_spmm_knobs_read = false; }

void spmm(torch::Tensor out, torch::Tensor x, torch::Tensor rowptr,
          torch::Tensor colidx, c10::optional<torch::Tensor> deg_dst,
          c10::optional<torch::Tensor> deg_src,
          c10::optional<torch::Tensor> row_order, bool accumulate,
          int64_t col_base, int64_t ncols) {
  ROC_CHECK_DEV_CONT(out);
  ROC_CHECK_DEV_CONT(x);
  ROC_CHECK_DEV_CONT(rowptr);
  ROC_CHECK_DEV_CONT(colidx);
  TORCH_CHECK(rowptr.scalar_type() == torch::kInt64, "rowptr must be int64");
  TORCH_CHECK(colidx.scalar_type() == torch::kInt32, "colidx must be int32");
  TORCH_CHECK(out.scalar_type() == x.scalar_type() ||
                  (out.scalar_type() == torch::kFloat32 &&
                   x.scalar_type() == torch::kBFloat16),
              "dtype mismatch (out must match x, or be fp32 for bf16 x)");
  const int num_rows = (int)out.size(0);
  const int64_t D = out.size(1);
  TORCH_CHECK(x.size(1) == D, "feature dim mismatch");
  TORCH_CHECK(rowptr.size(0) == num_rows + 1, "rowptr size mismatch");
  // column-phase pass: touch only [col_base, col_base+ncols) of every
  // row (ncols == 0 -> the whole row). The stride stays D.
  const int64_t col_end = ncols > 0 ? col_base + ncols : D;
  TORCH_CHECK(col_base >= 0 && col_base < col_end && col_end <= D,
              "bad column window");
  const float* dd =
      deg_dst.has_value() ? deg_dst->data_ptr<float>() : nullptr;
  const float* ds =
      deg_src.has_value() ? deg_src->data_ptr<float>() : nullptr;
  const int* ro =
      row_order.has_value() ? row_order->data_ptr<int>() : nullptr;
  auto stream = roc_stream();
  if (x.scalar_type() == torch::kBFloat16 &&
      out.scalar_type() == torch::kBFloat16) {
    launch_spmm<unsigned short, unsigned short>(
        (unsigned short*)out.data_ptr(), (const unsigned short*)x.data_ptr(),
        rowptr.data_ptr<int64_t>(), colidx.data_ptr<int>(), dd, ds, ro,
        num_rows, D, col_base, col_end, accumulate, (size_t)x.numel(),
        stream);
  } else if (x.scalar_type() == torch::kBFloat16) {
    launch_spmm<unsigned short, float>(
        out.data_ptr<float>(), (const unsigned short*)x.data_ptr(),
        rowptr.data_ptr<int64_t>(), colidx.data_ptr<int>(), dd, ds, ro,
        num_rows, D, col_base, col_end, accumulate, (size_t)x.numel(),
        stream);
  } else if (x.scalar_type() == torch::kFloat32) {
    launch_spmm<float, float>(out.data_ptr<float>(), x.data_ptr<float>(),
                       rowptr.data_ptr<int64_t>(), colidx.data_ptr<int>(), dd,
                       ds, ro, num_rows, D, col_base, col_end, accumulate,
                       (size_t)x.numel(), stream);
  } else {
    TORCH_CHECK(false, "spmm: unsupported dtype (bf16/f32 only)");
  }
  ROC_HIP_CHECK(hipGetLastError());
}

void spmm_edge(torch::Tensor out, torch::Tensor x, torch::Tensor rowptr,
               torch::Tensor colidx, torch::Tensor edge_val,
               c10::optional<torch::Tensor> deg_dst,
               c10::optional<torch::Tensor> row_order, bool accumulate) {
  ROC_CHECK_DEV_CONT(out);
  ROC_CHECK_DEV_CONT(x);
  ROC_CHECK_DEV_CONT(rowptr);
  ROC_CHECK_DEV_CONT(colidx);
  ROC_CHECK_DEV_CONT(edge_val);
  TORCH_CHECK(rowptr.scalar_type() == torch::kInt64, "rowptr must be int64");
  TORCH_CHECK(colidx.scalar_type() == torch::kInt32, "colidx must be int32");
  TORCH_CHECK(edge_val.scalar_type() == torch::kFloat32,
              "edge_val must be fp32");
  TORCH_CHECK(edge_val.numel() == colidx.numel(), "edge_val size mismatch");
  TORCH_CHECK(out.scalar_type() == x.scalar_type(), "dtype mismatch");
  const int num_rows = (int)out.size(0);
  const int64_t D = out.size(1);
  TORCH_CHECK(x.size(1) == D, "feature dim mismatch");
  TORCH_CHECK(rowptr.size(0) == num_rows + 1, "rowptr size mismatch");
  const float* dd =
      deg_dst.has_value() ? deg_dst->data_ptr<float>() : nullptr;
  const int* ro =
      row_order.has_value() ? row_order->data_ptr<int>() : nullptr;
  auto stream = roc_stream();
  if (x.scalar_type() == torch::kBFloat16) {
    launch_spmm_edge<unsigned short>(
        (unsigned short*)out.data_ptr(), (const unsigned short*)x.data_ptr(),
        rowptr.data_ptr<int64_t>(), colidx.data_ptr<int>(),
        edge_val.data_ptr<float>(), dd, ro, num_rows, D, accumulate,
        (size_t)x.numel(), stream);
  } else if (x.scalar_type() == torch::kFloat32) {
    launch_spmm_edge<float>(
        out.data_ptr<float>(), x.data_ptr<float>(),
        rowptr.data_ptr<int64_t>(), colidx.data_ptr<int>(),
        edge_val.data_ptr<float>(), dd, ro, num_rows, D, accumulate,
        (size_t)x.numel(), stream);
  } else {
    TORCH_CHECK(false, "spmm_edge: unsupported dtype (bf16/f32 only)");
  }
  ROC_HIP_CHECK(hipGetLastError());
}

void edge_dot(torch::Tensor dw, torch::Tensor dy, torch::Tensor x,
              torch::Tensor rowptr, torch::Tensor colidx) {
  ROC_CHECK_DEV_CONT(dw);
  ROC_CHECK_DEV_CONT(dy);
  ROC_CHECK_DEV_CONT(x);
  ROC_CHECK_DEV_CONT(rowptr);
  ROC_CHECK_DEV_CONT(colidx);
  TORCH_CHECK(dw.scalar_type() == torch::kFloat32, "dw must be fp32");
  TORCH_CHECK(dw.numel() == colidx.numel(), "dw size mismatch");
  TORCH_CHECK(dy.scalar_type() == x.scalar_type(), "dtype mismatch");
  const int num_rows = (int)dy.size(0);
  const int64_t D = dy.size(1);
  TORCH_CHECK(x.size(1) == D, "feature dim mismatch");
  TORCH_CHECK(rowptr.size(0) == num_rows + 1, "rowptr size mismatch");
  auto stream = roc_stream();
  if (dy.scalar_type() == torch::kBFloat16) {
    launch_edge_dot<unsigned short>(
        dw.data_ptr<float>(), (const unsigned short*)dy.data_ptr(),
        (const unsigned short*)x.data_ptr(), rowptr.data_ptr<int64_t>(),
        colidx.data_ptr<int>(), num_rows, D, stream);
  } else if (dy.scalar_type() == torch::kFloat32) {
    launch_edge_dot<float>(
        dw.data_ptr<float>(), dy.data_ptr<float>(), x.data_ptr<float>(),
        rowptr.data_ptr<int64_t>(), colidx.data_ptr<int>(), num_rows, D,
        stream);
  } else {
    TORCH_CHECK(false, "edge_dot: unsupported dtype (bf16/f32 only)");
  }
  ROC_HIP_CHECK(hipGetLastError());
}
