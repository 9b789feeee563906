"""Autograd operators with HIP-kernel dispatch.

Every op dispatches to the hand-written CDNA4 kernels in ``roc_amd._C``
when the tensors live on a GPU, and to the plain-PyTorch fp32 reference
(`reference.py`) on CPU. There is NO silent eager fallback on GPU: if the
extension is missing for a CUDA tensor, the op raises.

Op semantics follow the reference operator layer;
the architecture (halo-sharded CSR, bf16 activations + fp32 masters,
fused kernels) is MI355X-native.
"""
from __future__ import annotations

from typing import Optional

import torch

from . import reference as ref

try:
    from roc_amd import _C  # built by setup.py build_ext --inplace
except ImportError:  # pragma: no cover - exercised only when ext missing
    _C = None


def has_ext() -> bool:
    return _C is not None


_PHASE_COLS: Optional[int] = None

# -- async dW: the weight-grad GEMMs are independent of the big SpMM^T
# that follows them in backward, and gemm_atb ACCUMULATES (fp32
# atomics), so when the param's .grad buffer exists (flat-grad setup)
# the dW launches on a side stream and lands directly in .grad while
# the aggregation backward runs. dw_flush() joins the side stream
# before any gradient consumer (all-reduce / Adam).
# Default OFF by measurement (profiles/r29 addendum): the epoch is
# memory-saturated end-to-end, so overlapping the memory-bound dW with
# the memory-bound SpMM^T returns nothing (15.88 vs 15.83 ms sync) —
# the capability stays for compute-bound model mixes (ROC_ASYNC_DW=1).
_DW_STREAM = None
_DW_PENDING: list = []  # [(done_event, (x, dy) keep-alive refs)]
_ASYNC_DW: Optional[bool] = None


def _async_dw() -> bool:
    global _ASYNC_DW
    if _ASYNC_DW is None:
        import os
        _ASYNC_DW = os.environ.get("ROC_ASYNC_DW", "0") == "1"
    return _ASYNC_DW


def _dw_stream():
    global _DW_STREAM
    if _DW_STREAM is None:
        _DW_STREAM = torch.cuda.Stream()
    return _DW_STREAM


def dw_flush() -> None:
    """Make the current stream wait for every pending side-stream dW
    GEMM. Call before anything that reads or zeroes gradients."""
    if not _DW_PENDING:
        return
    from .. import streamcheck
    cs = torch.cuda.current_stream()
    for done, _refs in _DW_PENDING:
        streamcheck.consumer(done, "dw-consume")
        cs.wait_event(done)
    _DW_PENDING.clear()


def _phase_cols() -> int:
    """Column-phase width for strip-blocked SpMM passes (0 = off).

    Default OFF by measurement (profiles/r23): at equal gather
    working-set bytes, 64-col phases run the headline epoch 19.4 ms vs
    16.0 ms single-phase — narrow 128-B row gathers pay a ~20% service
    penalty vs whole 512-B rows, and the partial-traffic saving never
    lands because phases multiply the pass count right back.
    Pinned at first use for the same reason the kernel geometry knobs
    are: a per-launch getenv on the hot path silently desyncs from any
    hipGraph capture.
    """
    global _PHASE_COLS
    if _PHASE_COLS is None:
        import os
        _PHASE_COLS = int(os.environ.get("ROC_SPMM_PHASE_COLS", "0"))
    return _PHASE_COLS


def _hip(x: torch.Tensor) -> bool:
    if x.is_cuda:
        if _C is None:
            raise RuntimeError(
                "roc_amd._C extension is not built but a GPU tensor was passed. "
                "Run `python setup.py build_ext --inplace`. Refusing a silent "
                "eager fallback on GPU."
            )
        return True
    return False


# ---------------------------------------------------------------------------
# ScatterGather (CSR SpMM neighbor aggregation) — the hot op
# ---------------------------------------------------------------------------

class _SpMM(torch.autograd.Function):
    """out[v] = sum over in-neighbors of x[u] on a (possibly halo-extended,
    possibly degree-normalized) local CSR. Backward runs the same kernel on
    the transpose CSR (exact on asymmetric graphs)."""

    @staticmethod
    def forward(ctx, x, rowptr, colidx, t_rowptr, t_colidx, num_rows, num_ext,
                deg_dst, deg_src, row_order, t_row_order,
                fwd_strips=None, bwd_strips=None):
        ctx.save_for_backward(t_rowptr, t_colidx, deg_dst, deg_src, t_row_order)
        ctx.num_ext = num_ext
        ctx.bwd_strips = bwd_strips
        if _hip(x) and fwd_strips is not None and deg_src is None:
            # source-strip-blocked schedule: each pass gathers from an
            # L2-resident source window; partials accumulate in fp32
            # (one rounding total — same numerics as single-pass).
            # Wide feature dims additionally sweep column PHASES: a
            # 64-col phase over 4x-wider strips keeps the same gather
            # working set (rows x cols bytes) while the fp32 partial
            # buffer is re-read/re-written 4x less often — partial
            # traffic, not gathers, was the next bound (profiles/r23).
            out32 = torch.empty(num_rows, x.shape[1], dtype=torch.float32,
                                device=x.device)
            D = x.shape[1]
            ph = _phase_cols()
            if ph and D % ph == 0 and D // ph >= 2:
                for p in range(0, D, ph):
                    for i, (srp, sci) in enumerate(fwd_strips):
                        _C.spmm(out32, x, srp, sci, None, None, None,
                                i > 0, p, ph)
            else:
                for i, (srp, sci) in enumerate(fwd_strips):
                    _C.spmm(out32, x, srp, sci, None, None, None, i > 0)
            if x.dtype == torch.bfloat16 and x.shape[1] % 8 == 0:
                out = torch.empty(num_rows, x.shape[1],
                                  dtype=torch.bfloat16, device=x.device)
                _C.cast_rowscale(out, out32, deg_dst)  # fused epilogue
            else:
                if deg_dst is not None:
                    out32 = out32 * deg_dst.unsqueeze(1)
                out = out32.to(x.dtype)
        elif _hip(x):
            out = torch.empty(num_rows, x.shape[1], dtype=x.dtype, device=x.device)
            _C.spmm(out, x, rowptr, colidx, deg_dst, deg_src, row_order)
        else:
            xin = x
            if deg_src is not None:  # deg_* are precomputed row scale factors
                xin = xin * deg_src.unsqueeze(1).to(xin.dtype)
            out = ref.spmm(xin, rowptr, colidx, num_rows)
            if deg_dst is not None:
                out = out * deg_dst.unsqueeze(1).to(out.dtype)
        return out

    @staticmethod
    def backward(ctx, dy):
        t_rowptr, t_colidx, deg_dst, deg_src, t_row_order = ctx.saved_tensors
        dy = dy.contiguous()
        # d/dx of  diag(a) A diag(b) x  =  diag(b) A^T diag(a) dy
        if _hip(dy):
            if deg_dst is not None:
                # pre-scale the gradient rows ONCE (one cheap pass) instead
                # of a per-edge deg gather inside the transpose SpMM
                tmp = torch.empty_like(dy)
                _C.rowscale(tmp, dy, deg_dst)
                dy = tmp
            if ctx.bwd_strips is not None and deg_src is None:
                dx32 = torch.empty(ctx.num_ext, dy.shape[1],
                                   dtype=torch.float32, device=dy.device)
                D = dy.shape[1]
                ph = _phase_cols()
                if ph and D % ph == 0 and D // ph >= 2:
                    for p in range(0, D, ph):
                        for i, (srp, sci) in enumerate(ctx.bwd_strips):
                            _C.spmm(dx32, dy, srp, sci, None, None, None,
                                    i > 0, p, ph)
                else:
                    for i, (srp, sci) in enumerate(ctx.bwd_strips):
                        _C.spmm(dx32, dy, srp, sci, None, None, None, i > 0)
                if dy.dtype == torch.bfloat16 and dy.shape[1] % 8 == 0:
                    dx = torch.empty(ctx.num_ext, dy.shape[1],
                                     dtype=torch.bfloat16,
                                     device=dy.device)
                    _C.cast_rowscale(dx, dx32, None)
                    return (dx,) + (None,) * 12
                return (dx32.to(dy.dtype),) + (None,) * 12
            dx = torch.empty(ctx.num_ext, dy.shape[1], dtype=dy.dtype, device=dy.device)
            _C.spmm(dx, dy, t_rowptr, t_colidx, deg_src, None, t_row_order)
        else:
            yin = dy
            if deg_dst is not None:
                yin = yin * deg_dst.unsqueeze(1).to(yin.dtype)
            dx = ref.spmm(yin, t_rowptr, t_colidx, ctx.num_ext)
            if deg_src is not None:
                dx = dx * deg_src.unsqueeze(1).to(dx.dtype)
        return (dx,) + (None,) * 12


class _SpMMEdge(torch.autograd.Function):
    """out[v] = sum_{e in row v} w[e] * x[col_e]  (+ optional dst row
    scale). The edge-tensor consumer op: w is a per-local-edge tensor
    in CSR edge order (the reference declares edge tensors,
    `gnn.cc:475-623`, but ships no op over them — this makes the
    surface real). Backward: dx rides the transpose CSR with permuted
    weights; dw[e] = <dy[row_e], x[col_e]>."""

    @staticmethod
    def forward(ctx, x, w, rowptr, colidx, t_rowptr, t_colidx, t_eperm,
                num_rows, num_ext, dst_scale, row_order, t_row_order):
        w = w.contiguous()
        ctx.save_for_backward(x, w, rowptr, colidx, t_rowptr, t_colidx,
                              t_eperm, dst_scale, t_row_order)
        ctx.num_ext = num_ext
        if _hip(x):
            out = torch.empty(num_rows, x.shape[1], dtype=x.dtype,
                              device=x.device)
            _C.spmm_edge(out, x, rowptr, colidx, w, dst_scale, row_order,
                         False)
        else:
            out = ref.spmm_weighted(x, rowptr, colidx, w, num_rows)
            if dst_scale is not None:
                out = out * dst_scale.unsqueeze(1).to(out.dtype)
        return out

    @staticmethod
    def backward(ctx, dy):
        (x, w, rowptr, colidx, t_rowptr, t_colidx, t_eperm, dst_scale,
         t_row_order) = ctx.saved_tensors
        dy = dy.contiguous()
        need_x, need_w = ctx.needs_input_grad[0], ctx.needs_input_grad[1]
        dx = dw = None
        if _hip(dy):
            if dst_scale is not None:
                tmp = torch.empty_like(dy)
                _C.rowscale(tmp, dy, dst_scale)
                dy = tmp
            if need_x:
                t_w = w[t_eperm].contiguous()
                dx = torch.empty(ctx.num_ext, dy.shape[1], dtype=dy.dtype,
                                 device=dy.device)
                _C.spmm_edge(dx, dy, t_rowptr, t_colidx, t_w, None,
                             t_row_order, False)
            if need_w:
                dw = torch.empty_like(w)
                _C.edge_dot(dw, dy, x, rowptr, colidx)
        else:
            if dst_scale is not None:
                dy = dy * dst_scale.unsqueeze(1).to(dy.dtype)
            if need_x:
                dx = ref.spmm_weighted(dy, t_rowptr, t_colidx, w[t_eperm],
                                       ctx.num_ext)
            if need_w:
                dw = ref.edge_dot(dy, x, rowptr, colidx).to(w.dtype)
        return (dx, dw) + (None,) * 10


def scatter_gather_weighted(x, w, shard, dst_scale=None):
    """Per-edge-weighted neighbor aggregation over the shard's local CSR.

    x: [n_ext, D] halo-extended features; w: fp32 [num_local_edges] in
    CSR edge order (see roc_amd.edge_tensor). Returns [n_local, D].
    Both x and w receive gradients. Works at any world size — edge
    tensors are partition-local, so no extra communication beyond the
    usual halo exchange of x."""
    return _SpMMEdge.apply(
        x, w, shard.rowptr, shard.colidx, shard.t_rowptr, shard.t_colidx,
        shard.t_edge_perm(), shard.n_local, shard.n_ext, dst_scale,
        shard.row_order, shard.t_row_order)


class _EdgeSoftmax(torch.autograd.Function):
    """Per-destination-row segment softmax over edge scores (the GAT
    attention normalizer). GPU: one fused CDNA4 kernel per direction
    (edge_softmax.hip); CPU: index-op reference."""

    @staticmethod
    def forward(ctx, s, rowptr, row_of_edge):
        s = s.contiguous()
        if _hip(s):
            alpha = torch.empty_like(s)
            _C.edge_softmax_fwd(alpha, s, rowptr)
        else:
            n = rowptr.numel() - 1
            m = torch.full((n,), float("-inf"), dtype=s.dtype)
            m = m.scatter_reduce(0, row_of_edge, s, reduce="amax",
                                 include_self=True)
            ex = (s - m[row_of_edge]).exp()
            denom = torch.zeros(n, dtype=s.dtype).index_add_(
                0, row_of_edge, ex)
            alpha = ex / denom[row_of_edge]
        ctx.save_for_backward(alpha, rowptr, row_of_edge)
        return alpha

    @staticmethod
    def backward(ctx, dalpha):
        alpha, rowptr, row_of_edge = ctx.saved_tensors
        dalpha = dalpha.contiguous()
        if _hip(dalpha):
            ds = torch.empty_like(dalpha)
            _C.edge_softmax_bwd(ds, dalpha, alpha, rowptr)
        else:
            n = rowptr.numel() - 1
            dot = torch.zeros(n, dtype=alpha.dtype).index_add_(
                0, row_of_edge, alpha * dalpha)
            ds = alpha * (dalpha - dot[row_of_edge])
        return ds, None, None


def edge_softmax(scores, shard):
    """softmax of fp32 edge scores within each destination row's
    in-edge segment: Σ_{e∈row v} out[e] = 1. Differentiable; pairs
    with scatter_gather_weighted for attention-style aggregation."""
    return _EdgeSoftmax.apply(scores.float(), shard.rowptr,
                              shard.row_of_edge())


class _AttentionSoftmax(torch.autograd.Function):
    """alpha[e] = softmax_row(lrelu(s_src[col_e] + s_dst[row])) — the
    whole GAT score+normalize chain in ONE kernel per direction
    (edge_softmax.hip att_*). Replaces 4 E-length torch intermediates
    and a 115M-element index_add backward. CPU path composes index
    ops (the numerics oracle)."""

    @staticmethod
    def forward(ctx, s_src, s_dst, rowptr, colidx, row_of_edge, slope):
        s_src = s_src.contiguous()
        s_dst = s_dst.contiguous()
        if _hip(s_src):
            alpha = torch.empty(colidx.numel(), dtype=torch.float32,
                                device=s_src.device)
            _C.att_softmax_fwd(alpha, s_src, s_dst, rowptr, colidx, slope)
        else:
            n = rowptr.numel() - 1
            sc = s_src[colidx.long()] + s_dst[row_of_edge]
            sc = torch.nn.functional.leaky_relu(sc, slope)
            m = torch.full((n,), float("-inf"), dtype=sc.dtype)
            m = m.scatter_reduce(0, row_of_edge, sc, reduce="amax",
                                 include_self=True)
            ex = (sc - m[row_of_edge]).exp()
            den = torch.zeros(n, dtype=sc.dtype).index_add_(
                0, row_of_edge, ex)
            alpha = ex / den[row_of_edge]
        ctx.save_for_backward(alpha, s_src, s_dst, rowptr, colidx,
                              row_of_edge)
        ctx.slope = slope
        return alpha

    @staticmethod
    def backward(ctx, dalpha):
        alpha, s_src, s_dst, rowptr, colidx, row_of_edge = ctx.saved_tensors
        slope = ctx.slope
        dalpha = dalpha.contiguous()
        if _hip(dalpha):
            dsrc = torch.zeros_like(s_src)
            dsdst = torch.empty_like(s_dst)
            _C.att_softmax_bwd(dsrc, dsdst, dalpha, alpha, s_src, s_dst,
                               rowptr, colidx, slope)
        else:
            n = rowptr.numel() - 1
            dot = torch.zeros(n, dtype=alpha.dtype).index_add_(
                0, row_of_edge, alpha * dalpha)
            ds = alpha * (dalpha - dot[row_of_edge])
            raw = s_src[colidx.long()] + s_dst[row_of_edge]
            dscore = torch.where(raw > 0, ds, slope * ds)
            dsrc = torch.zeros_like(s_src).index_add_(
                0, colidx.long(), dscore)
            dsdst = torch.zeros_like(s_dst).index_add_(
                0, row_of_edge, dscore)
        return dsrc, dsdst, None, None, None, None


def attention_softmax(s_src, s_dst, shard, slope: float = 0.2):
    """Fused GAT attention coefficients from per-node score halves:
    s_src fp32 [n_ext] (source half, halo-extended), s_dst fp32
    [n_local]. Returns alpha fp32 [E_local]; both inputs get grads."""
    return _AttentionSoftmax.apply(s_src.float(), s_dst.float(),
                                   shard.rowptr, shard.colidx,
                                   shard.row_of_edge(), slope)


def scatter_gather(x, shard, normalize: bool = False, dst_scale=None,
                   src_scale=None):
    """Neighbor sum-aggregation over the shard's local CSR.

    x: [n_ext, D] halo-extended features. Returns [n_local, D].
    normalize=True fuses the symmetric D^-1/2 A D^-1/2 GCN normalization
    (the reference composes indegree_norm -> scatter_gather -> indegree_norm,
    `gnn.cc:82-84`; the fused kernel reads each feature once instead).
    dst_scale/src_scale: explicit per-row fp32 factors. The fast GCN path
    pre-scales sources in the linear epilogue and passes only dst_scale
    here (saves one gather per edge).
    """
    deg_dst = shard.rsqrt_deg_local if normalize else dst_scale
    deg_src = shard.rsqrt_deg_ext if normalize else src_scale
    return _SpMM.apply(
        x, shard.rowptr, shard.colidx, shard.t_rowptr, shard.t_colidx,
        shard.n_local, shard.n_ext, deg_dst, deg_src,
        shard.row_order, shard.t_row_order,
        shard.fwd_strips, shard.bwd_strips,
    )


# ---------------------------------------------------------------------------
# InDegreeNorm
# ---------------------------------------------------------------------------

class _DegScale(torch.autograd.Function):
    """out = x * scale_row (rowwise). Self-adjoint: backward applies the
    same scaling to the gradient (reference `graphnorm_kernel.cu:126-136`)."""

    @staticmethod
    def forward(ctx, x, scale):
        ctx.save_for_backward(scale)
        if _hip(x):
            out = torch.empty_like(x)
            _C.rowscale(out, x, scale)
        else:
            out = x * scale.unsqueeze(1).to(x.dtype)
        return out

    @staticmethod
    def backward(ctx, dy):
        (scale,) = ctx.saved_tensors
        dy = dy.contiguous()
        if _hip(dy):
            dx = torch.empty_like(dy)
            _C.rowscale(dx, dy, scale)
        else:
            dx = dy * scale.unsqueeze(1).to(dy.dtype)
        return dx, None


def indegree_norm(x, shard):
    """out[v] = x[v] / sqrt(indeg(v)) on local rows (reference graphnorm op)."""
    return _DegScale.apply(x, shard.rsqrt_deg_local)


def degree_scale(x, scale):
    """Generic rowwise scale (used by GraphSAGE mean aggregation: 1/deg)."""
    return _DegScale.apply(x, scale)


# ---------------------------------------------------------------------------
# Linear (hand-written MFMA GEMM on GPU)
# ---------------------------------------------------------------------------

class _Linear(torch.autograd.Function):
    """y = x @ w (+ fused ReLU). w is an fp32 master [in_dim, out_dim];
    on GPU x is bf16 and w is cast per call (w is tiny: <1 MB).

    Backward: dw = x^T dy (fp32 accumulate, split-K), dx = dy @ w^T.
    Reference: `linear_kernel.cu:20-127` (fwd+fused relu), `:130-245` (bwd).
    """

    @staticmethod
    def forward(ctx, x, w, act, row_scale):
        if _hip(x):
            # gemm_rr takes B pre-transposed: Bt = w^T [out,in]
            wt, wc = _cast_weight(w, x.dtype)
            ctx.wc = wc
            y = torch.empty(x.shape[0], w.shape[1], dtype=x.dtype, device=x.device)
            _C.gemm_rr(y, x, wt, act == "relu", row_scale)
            if act == "sigmoid":
                _C.sigmoid_fwd(y, y)
        else:
            y = ref.linear(x, w)
            if row_scale is not None:
                y = y * row_scale.unsqueeze(1).to(y.dtype)
            if act == "relu":
                y = ref.relu(y)
            elif act == "sigmoid":
                y = ref.sigmoid(y)
        ctx.save_for_backward(x, w, y if act else None, row_scale)
        ctx.act = act
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, y, row_scale = ctx.saved_tensors
        act = ctx.act
        dy = dy.contiguous()
        if _hip(dy):
            if act == "relu":
                dym = torch.empty_like(dy)
                _C.relu_bwd(dym, dy, y)
                dy = dym
            elif act == "sigmoid":
                dym = torch.empty_like(dy)
                _C.sigmoid_bwd(dym, dy, y)
                dy = dym
            if row_scale is not None:
                dym = torch.empty_like(dy)
                _C.rowscale(dym, dy, row_scale)
                dy = dym
            gbuf = w.grad if _async_dw() else None
            if (gbuf is not None and gbuf.dtype == torch.float32
                    and gbuf.is_contiguous() and gbuf.is_cuda):
                # accumulate straight into .grad on the side stream
                # (gemm_atb is += via fp32 atomics); return None so
                # autograd does not accumulate a second copy
                from .. import streamcheck
                s = _dw_stream()
                ready = torch.cuda.Event()
                ready.record()  # x/dy produced on the current stream
                s.wait_event(ready)
                with torch.cuda.stream(s):
                    _C.gemm_atb(gbuf, x, dy)
                done = torch.cuda.Event()
                done.record(s)
                streamcheck.producer(done, "dw-gemm")
                _DW_PENDING.append((done, (x, dy)))
                dw = None
            else:
                dw = torch.zeros_like(w)  # fp32 [in, out]
                _C.gemm_atb(dw, x, dy)
            dx = None
            if ctx.needs_input_grad[0]:  # layer-1 inputs carry no grad
                # dx = dy @ w^T: gemm_rr's Bt = (w^T)^T = w [in,out]
                dx = torch.empty_like(x)
                _C.gemm_rr(dx, dy, ctx.wc, False)
        else:
            if act == "relu":
                dy = ref.relu_grad(dy, y)
            elif act == "sigmoid":
                dy = ref.sigmoid_grad(dy, y)
            if row_scale is not None:
                dy = dy * row_scale.unsqueeze(1).to(dy.dtype)
            dw = (x.to(torch.float32).t() @ dy.to(torch.float32))
            dx = (dy @ w.t().to(dy.dtype)) if ctx.needs_input_grad[0] else None
        return dx, dw, None, None


def linear(x, w, activation: Optional[str] = None, row_scale=None):
    """y = act((x @ w) * row_scale). row_scale rides the GEMM epilogue
    (used for the GCN source-degree pre-normalization)."""
    return _Linear.apply(x, w, activation, row_scale)


# ---------------------------------------------------------------------------
# Standalone activations (reference activation op)
# ---------------------------------------------------------------------------

class _Act(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, kind):
        if _hip(x):
            y = torch.empty_like(x)
            getattr(_C, f"{kind}_fwd")(y, x)
        else:
            y = ref.relu(x) if kind == "relu" else ref.sigmoid(x)
        ctx.save_for_backward(y)
        ctx.kind = kind
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        dy = dy.contiguous()
        if _hip(dy):
            dx = torch.empty_like(dy)
            getattr(_C, f"{ctx.kind}_bwd")(dx, dy, y)
        else:
            dx = ref.relu_grad(dy, y) if ctx.kind == "relu" else ref.sigmoid_grad(dy, y)
        return dx, None


def relu(x):
    return _Act.apply(x, "relu")


def sigmoid(x):
    return _Act.apply(x, "sigmoid")


# ---------------------------------------------------------------------------
# Element (add / mul) — reference element op; autograd-composable
# ---------------------------------------------------------------------------

class _Add(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, b):
        if _hip(a):
            out = torch.empty_like(a)
            _C.ewise_add(out, a, b)
            return out
        return a + b

    @staticmethod
    def backward(ctx, dy):
        return dy, dy


class _Mul(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, b):
        ctx.save_for_backward(a, b)
        if _hip(a):
            out = torch.empty_like(a)
            _C.ewise_mul(out, a, b)
            return out
        return a * b

    @staticmethod
    def backward(ctx, dy):
        a, b = ctx.saved_tensors
        dy = dy.contiguous()
        if _hip(dy):
            da = torch.empty_like(a)
            db = torch.empty_like(b)
            _C.ewise_mul(da, dy, b)
            _C.ewise_mul(db, dy, a)
            return da, db
        return dy * b, dy * a


def add(a, b):
    return _Add.apply(a, b)


def mul(a, b):
    return _Mul.apply(a, b)


# ---------------------------------------------------------------------------
# Dropout (Philox counter-based; mask regenerated in backward — no mask
# tensor is ever materialized on GPU)
# ---------------------------------------------------------------------------

_DROPOUT_STATE = {"seed": 1, "offset": 0, "counter": None, "epoch": 0}
_WEIGHT_VERSION = [0]


def bump_weight_version() -> None:
    """Called by the optimizer after a step: invalidates cached weight
    casts (the per-call w -> bf16 (+transpose) casts are reused within a
    step; disabled under hipGraph capture where the cast kernels must be
    part of the replayed graph)."""
    _WEIGHT_VERSION[0] += 1


def _cast_weight(w, dtype):
    if _DROPOUT_STATE["counter"] is not None:  # capture mode: no caching
        wc = w.contiguous().to(dtype)
        return wc.t().contiguous(), wc
    cache = getattr(w, "_roc_cast", None)
    ver = _WEIGHT_VERSION[0]
    if cache is not None and cache[0] == ver and cache[2].dtype == dtype:
        return cache[1], cache[2]
    wc = w.contiguous().to(dtype)
    wt = wc.t().contiguous()
    w._roc_cast = (ver, wt, wc)
    return wt, wc


def set_dropout_seed(seed: int) -> None:
    _DROPOUT_STATE["seed"] = int(seed)
    _DROPOUT_STATE["offset"] = 0
    _DROPOUT_STATE["epoch"] = 0


def set_dropout_counter(counter) -> None:
    """Device int64[1] epoch counter: makes dropout hipGraph-replayable
    (offset = counter * 65536 + per-epoch call index)."""
    _DROPOUT_STATE["counter"] = counter
    _DROPOUT_STATE["offset"] = 0


def reset_dropout_offset() -> None:
    _DROPOUT_STATE["offset"] = 0


def next_dropout_epoch() -> None:
    """Advance the per-epoch Philox base for call_id-addressed dropout
    (eager path; the hipGraph path gets its epoch from the device
    counter instead). Masks then vary per epoch while staying exactly
    reproducible WITHIN the epoch — the property activation recompute
    needs."""
    _DROPOUT_STATE["epoch"] += 1


class _Dropout(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, p, call_id=None):
        seed = _DROPOUT_STATE["seed"]
        counter = _DROPOUT_STATE["counter"]
        if call_id is not None:
            # stable per-epoch offset: a checkpointed layer re-running its
            # forward during backward regenerates the IDENTICAL mask.
            # Epoch variation: device counter (graph mode) or host epoch.
            offset = int(call_id)
            if counter is None:
                offset += _DROPOUT_STATE["epoch"] * 65536
        else:
            offset = _DROPOUT_STATE["offset"]
            _DROPOUT_STATE["offset"] += 1
        ctx.params = (p, seed, offset, counter)
        if _hip(x):
            y = torch.empty_like(x)
            _C.dropout_fwd(y, x, p, seed, offset, counter)
            return y
        mask = ref.dropout_mask(x.shape, p, seed, offset, x.device)
        ctx.save_for_backward(mask)
        return x * mask.to(x.dtype) / (1.0 - p)

    @staticmethod
    def backward(ctx, dy):
        p, seed, offset, counter = ctx.params
        dy = dy.contiguous()
        if _hip(dy):
            dx = torch.empty_like(dy)
            _C.dropout_fwd(dx, dy, p, seed, offset, counter)  # same stream
            return dx, None, None
        (mask,) = ctx.saved_tensors
        return dy * mask.to(dy.dtype) / (1.0 - p), None, None


def dropout(x, p: float, training: bool, call_id=None):
    """Train: mask+scale; infer: identity (reference `dropout_kernel.cu:159-180`).

    ``call_id``: stable per-epoch Philox offset (= the model's layer
    index). Required under activation recompute; numerically identical
    to the global running offset for the shipped models (one dropout
    per layer, called in layer order)."""
    if not training or p <= 0.0:
        return x
    return _Dropout.apply(x, p, call_id)


# ---------------------------------------------------------------------------
# SoftmaxCrossEntropy (fused loss + grad + metrics)
# ---------------------------------------------------------------------------

class _SoftmaxCE(torch.autograd.Function):
    """Returns (loss, metrics[8]) where metrics =
    [roc_loss_sum, ce_loss_sum, train_correct, train_total,
     val_correct, val_total, test_correct, test_total] (float32, on device).
    loss = ce_loss_sum (differentiable driver for .backward()).
    Gradient: (softmax - onehot) * (mask == Train) * grad_scale
    (reference `softmax_kernel.cu:19-79`).
    """

    @staticmethod
    def forward(ctx, logits, labels, mask, grad_scale, num_classes):
        if _hip(logits):
            dl = torch.empty_like(logits)
            metrics = torch.zeros(8, dtype=torch.float32, device=logits.device)
            _C.softmax_ce(dl, metrics, logits, labels, mask, grad_scale,
                          num_classes or -1)
            loss = metrics[1].clone()
        else:
            dl, md = ref.softmax_cross_entropy(logits, labels, mask,
                                               grad_scale, num_classes)
            metrics = torch.tensor(
                [md["roc_loss"], md["ce_loss"] * max(md["train_total"], 1),
                 md["train_correct"], md["train_total"],
                 md["val_correct"], md["val_total"],
                 md["test_correct"], md["test_total"]],
                dtype=torch.float32)
            # differentiable loss keeps the LOGITS dtype (an fp32 round-trip
            # through the metrics container breaks fp64 gradcheck)
            loss = torch.tensor(md["ce_loss"] * max(md["train_total"], 1),
                                dtype=logits.dtype)
        ctx.save_for_backward(dl)
        ctx.mark_non_differentiable(metrics)
        return loss, metrics

    @staticmethod
    def backward(ctx, dloss, _dmetrics):
        (dl,) = ctx.saved_tensors
        return dl * dloss, None, None, None, None


def softmax_cross_entropy(logits, labels, mask, grad_scale: float = 1.0,
                          num_classes: Optional[int] = None):
    """num_classes < logits width runs softmax over the first num_classes
    columns only (class-dim padding for 16-B-aligned rows)."""
    return _SoftmaxCE.apply(logits, labels, mask, grad_scale, num_classes)


def decode_metrics(metrics: torch.Tensor) -> dict:
    m = metrics.cpu().tolist()
    tt = max(m[3], 1.0)
    return {
        "roc_loss": m[0],
        "ce_loss": m[1] / tt,
        "train_acc": m[2] / tt,
        "val_acc": m[4] / max(m[5], 1.0),
        "test_acc": m[6] / max(m[7], 1.0),
        "train_total": int(m[3]),
        "val_total": int(m[5]),
        "test_total": int(m[7]),
    }


# ---------------------------------------------------------------------------
# Adam (not autograd — called by the optimizer)
# ---------------------------------------------------------------------------

def adam_step(w, g, m, v, alpha_t, beta1, beta2, eps, weight_decay,
              step=None, decay_rate=1.0, decay_steps=100):
    """step (device int64[1]) switches to the hipGraph-replayable mode:
    alpha_t is then the BASE lr and the bias-corrected decayed step size
    is derived on device from the counter."""
    if _hip(w):
        _C.adam_step(w, g, m, v, alpha_t, beta1, beta2, eps, weight_decay,
                     step, decay_rate, decay_steps)
    else:
        ref.adam_step(w, g, m, v, alpha_t, beta1, beta2, eps, weight_decay)
