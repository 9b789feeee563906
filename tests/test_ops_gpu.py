"""GPU kernel numerics: every CDNA4 kernel vs the plain-PyTorch fp32
reference of the same op (bf16 paths compare against fp32 references
computed FROM the bf16-rounded inputs, so only accumulation error
remains)."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu

from roc_amd.graph import synthetic_graph, MASK_TRAIN
from roc_amd.parallel.partition import build_shard
from roc_amd.ops import functional as F
from roc_amd.ops import reference as ref

DEV = "cuda:0"


def _ext():
    from roc_amd import _C
    return _C


@pytest.fixture(scope="module")
def graph_small():
    g = synthetic_graph(3000, 60000, seed=5)
    shard = build_shard(g, 0, 1)
    return g, shard


# ---------------------------------------------------------------------------
# SpMM
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("D", [256, 48, 41, 602])
def test_spmm_fp32_matches_cpu(graph_small, D):
    g, shard = graph_small
    torch.manual_seed(0)
    x = torch.randn(g.num_nodes, D)
    want = ref.spmm(x, g.rowptr, g.colidx, g.num_nodes)
    shd = shard.to(DEV)
    got = F.scatter_gather(x.to(DEV), shd).cpu()
    assert torch.allclose(got, want, rtol=1e-4, atol=1e-3), \
        (got - want).abs().max()


@pytest.mark.parametrize("D", [256, 41])
def test_spmm_bf16(graph_small, D):
    g, shard = graph_small
    torch.manual_seed(1)
    x = torch.randn(g.num_nodes, D).to(torch.bfloat16)
    want = ref.spmm(x.float(), g.rowptr, g.colidx, g.num_nodes)
    shd = shard.to(DEV)
    got = F.scatter_gather(x.to(DEV), shd).cpu().float()
    # bf16 output rounding of fp32 accumulation
    tol = want.abs().max().item() * 2 ** -7
    assert torch.allclose(got, want, atol=tol, rtol=0.02), \
        (got - want).abs().max()


def test_spmm_fused_norm(graph_small):
    g, shard = graph_small
    torch.manual_seed(2)
    x = torch.randn(g.num_nodes, 64)
    shd = shard.to(DEV)
    got = F.scatter_gather(x.to(DEV), shd, normalize=True).cpu()
    want = F.scatter_gather(x, shard, normalize=True)
    assert torch.allclose(got, want, rtol=1e-3, atol=1e-4), \
        (got - want).abs().max()


def test_spmm_backward_gpu(graph_small):
    g, shard = graph_small
    torch.manual_seed(3)
    shd = shard.to(DEV)
    x = torch.randn(g.num_nodes, 32, device=DEV, requires_grad=True)
    out = F.scatter_gather(x, shd)
    gy = torch.randn_like(out)
    out.backward(gy)
    xc = x.detach().cpu().clone().requires_grad_(True)
    out_c = F.scatter_gather(xc, shard)
    out_c.backward(gy.cpu())
    assert torch.allclose(x.grad.cpu(), xc.grad, rtol=1e-4, atol=1e-3)


# ---------------------------------------------------------------------------
# GEMM (MFMA)
# ---------------------------------------------------------------------------

def test_gemm_rr_identity_asymmetric():
    # A = I: C must equal B exactly (transpose-detecting per CDNA guide §3)
    n = 64
    A = torch.eye(n, dtype=torch.bfloat16, device=DEV)
    B = torch.arange(n * 48, dtype=torch.float32, device=DEV).reshape(n, 48)
    B = ((B % 13) - 6).to(torch.bfloat16)  # asymmetric, small ints (exact)
    C = torch.empty(n, 48, dtype=torch.bfloat16, device=DEV)
    _ext().gemm_rr(C, A, B.t().contiguous(), False)
    assert torch.equal(C, B), (C.float() - B.float()).abs().max()


@pytest.mark.parametrize("M,K,N", [(300, 608, 256), (257, 256, 41),
                                   (128, 41, 602), (1000, 256, 128),
                                   (64, 32, 64)])
def test_gemm_rr_random(M, K, N):
    torch.manual_seed(4)
    A = torch.randn(M, K).to(torch.bfloat16)
    B = torch.randn(K, N).to(torch.bfloat16)
    want = A.float() @ B.float()
    C = torch.empty(M, N, dtype=torch.bfloat16, device=DEV)
    _ext().gemm_rr(C, A.to(DEV), B.t().contiguous().to(DEV), False)
    got = C.cpu().float()
    tol = want.abs().max().item() * 2 ** -7 + 1e-3
    assert torch.allclose(got, want, atol=tol, rtol=0.05), \
        (got - want).abs().max()


@pytest.mark.parametrize("M,K,N", [(300, 608, 256), (257, 256, 41),
                                   (128, 33, 96)])
def test_gemm_rr_fp32(M, K, N):
    """exact-fp32 MFMA path (mfma_f32_16x16x4f32) vs torch fp32 matmul."""
    torch.manual_seed(12)
    A = torch.randn(M, K)
    B = torch.randn(K, N)
    want = A @ B
    C = torch.empty(M, N, dtype=torch.float32, device=DEV)
    _ext().gemm_rr(C, A.to(DEV), B.t().contiguous().to(DEV), False)
    got = C.cpu()
    # fp32 MFMA is an exact fmaf chain; only summation order differs
    tol = want.abs().max().item() * 1e-5 + 1e-4
    assert torch.allclose(got, want, atol=tol, rtol=1e-4), \
        (got - want).abs().max()


def test_gemm_atb_fp32():
    torch.manual_seed(13)
    A = torch.randn(2000, 96)
    B = torch.randn(2000, 64) * 0.1
    want = A.t() @ B
    C = torch.zeros(96, 64, dtype=torch.float32, device=DEV)
    _ext().gemm_atb(C, A.to(DEV), B.to(DEV))
    tol = want.abs().max().item() * 1e-5 + 1e-3
    assert torch.allclose(C.cpu(), want, atol=tol, rtol=1e-4)


def test_gemm_rr_fused_relu():
    torch.manual_seed(5)
    A = torch.randn(200, 64).to(torch.bfloat16)
    B = torch.randn(64, 96).to(torch.bfloat16)
    want = torch.relu(A.float() @ B.float())
    C = torch.empty(200, 96, dtype=torch.bfloat16, device=DEV)
    _ext().gemm_rr(C, A.to(DEV), B.t().contiguous().to(DEV), True)
    got = C.cpu().float()
    tol = want.abs().max().item() * 2 ** -7 + 1e-3
    assert torch.allclose(got, want, atol=tol, rtol=0.05)


@pytest.mark.parametrize("R,Ka,N", [(5000, 608, 256), (3000, 256, 41),
                                    (1000, 41, 64),
                                    # wide 128x128-tile path, ragged edges
                                    (2000, 200, 136), (1500, 136, 129)])
def test_gemm_atb(R, Ka, N):
    torch.manual_seed(6)
    A = torch.randn(R, Ka).to(torch.bfloat16)
    B = (torch.randn(R, N) * 0.1).to(torch.bfloat16)
    want = A.float().t() @ B.float()
    C = torch.zeros(Ka, N, dtype=torch.float32, device=DEV)
    _ext().gemm_atb(C, A.to(DEV), B.to(DEV))
    got = C.cpu()
    tol = want.abs().max().item() * 2 ** -7 + 1e-2
    assert torch.allclose(got, want, atol=tol, rtol=0.05), \
        (got - want).abs().max()


def test_linear_autograd_gpu():
    """Wiring test: backward grads must match a fp32 reference built from
    the GPU's OWN bf16 forward output (isolates kernel wiring from
    cross-precision relu-boundary divergence)."""
    torch.manual_seed(7)
    x = torch.randn(500, 64, dtype=torch.bfloat16, device=DEV,
                    requires_grad=True)
    w = torch.nn.Parameter(torch.randn(64, 32, device=DEV))
    y = F.linear(x, w, activation="relu")
    gy = (torch.randn_like(y.float())).to(torch.bfloat16)
    y.backward(gy)
    # reference from GPU intermediates
    y_c = y.detach().float().cpu()
    w_bf = w.detach().to(torch.bfloat16).float().cpu()
    x_c = x.detach().float().cpu()
    want_y = torch.relu(x_c @ w_bf)
    tol_y = want_y.abs().max().item() * 2 ** -7 + 1e-2
    assert torch.allclose(y_c, want_y, atol=tol_y, rtol=0.05)
    dy = gy.float().cpu() * (y_c > 0)  # exact: relu_bwd is a masked copy
    dw_want = x_c.t() @ dy
    tol_w = dw_want.abs().max().item() * 2 ** -7 + 1e-2
    assert torch.allclose(w.grad.cpu(), dw_want, atol=tol_w, rtol=0.05), \
        (w.grad.cpu() - dw_want).abs().max()
    dx_want = dy @ w_bf.t()
    tol_x = dx_want.abs().max().item() * 2 ** -7 + 1e-2
    assert torch.allclose(x.grad.float().cpu(), dx_want, atol=tol_x,
                          rtol=0.05), (x.grad.float().cpu() - dx_want).abs().max()


# ---------------------------------------------------------------------------
# elementwise / rowscale / dropout / softmax / adam
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_elementwise_gpu(dtype):
    torch.manual_seed(8)
    a = torch.randn(999, 7).to(dtype).to(DEV)
    b = torch.randn(999, 7).to(dtype).to(DEV)
    out = torch.empty_like(a)
    _ext().ewise_add(out, a, b)
    assert torch.allclose(out.float().cpu(), (a + b).float().cpu(), atol=1e-2)
    _ext().ewise_mul(out, a, b)
    assert torch.allclose(out.float().cpu(), (a * b).float().cpu(), atol=1e-2)
    _ext().relu_fwd(out, a)
    assert torch.equal(out.cpu(), torch.relu(a).cpu())
    _ext().sigmoid_fwd(out, a)
    assert torch.allclose(out.float().cpu(), torch.sigmoid(a.float()).cpu(),
                          atol=1e-2)


def test_rowscale_gpu():
    x = torch.randn(123, 37, device=DEV)
    s = torch.rand(123, device=DEV) + 0.5
    out = torch.empty_like(x)
    _ext().rowscale(out, x, s)
    assert torch.allclose(out.cpu(), (x * s.unsqueeze(1)).cpu(), atol=1e-6)


def test_dropout_gpu():
    x = torch.ones(100000, device=DEV, dtype=torch.bfloat16).reshape(1000, 100)
    y = torch.empty_like(x)
    _ext().dropout_fwd(y, x, 0.5, 1234, 7)
    kept = (y != 0).float().mean().item()
    assert 0.48 < kept < 0.52
    nz = y[y != 0].float()
    assert torch.allclose(nz, torch.full_like(nz, 2.0))
    # deterministic: same (seed, offset) -> same mask
    y2 = torch.empty_like(x)
    _ext().dropout_fwd(y2, x, 0.5, 1234, 7)
    assert torch.equal(y, y2)
    # different offset -> different mask
    _ext().dropout_fwd(y2, x, 0.5, 1234, 8)
    assert not torch.equal(y, y2)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_softmax_ce_gpu(dtype):
    torch.manual_seed(9)
    n, c = 2000, 41
    logits = torch.randn(n, c).to(dtype)
    labels = torch.randint(0, c, (n,))
    mask = torch.randint(1, 4, (n,), dtype=torch.int32)
    loss, metrics = F.softmax_cross_entropy(
        logits.to(DEV), labels.to(DEV), mask.to(DEV))
    dl_want, md_want = ref.softmax_cross_entropy(logits.float(), labels, mask)
    md = F.decode_metrics(metrics)
    assert md["train_total"] == md_want["train_total"]
    assert md["train_acc"] == pytest.approx(
        md_want["train_correct"] / md_want["train_total"], abs=1e-3)
    assert md["roc_loss"] == pytest.approx(md_want["roc_loss"], rel=3e-2)
    assert md["ce_loss"] == pytest.approx(
        md_want["ce_loss"], rel=3e-2)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_spmm_edge_gpu(dtype):
    """Edge-weighted SpMM fwd + both backward kernels (spmm_edge on the
    transpose, edge_dot) vs the fp32 CPU reference."""
    from roc_amd import build_shard, edge_tensor
    from roc_amd.graph import synthetic_dataset
    torch.manual_seed(19)
    g, feats, _, _, _ = synthetic_dataset("cora", scale=0.1, seed=5)
    sh0 = build_shard(g, 0, 1)
    D = 64
    x_cpu = torch.randn(g.num_nodes, D)
    w_cpu = torch.rand(sh0.num_local_edges)
    # CPU fp32 reference (autograd through the reference path)
    xc = x_cpu.clone().requires_grad_(True)
    wc = w_cpu.clone().requires_grad_(True)
    out_c = F.scatter_gather_weighted(xc, wc, sh0,
                                      dst_scale=sh0.rsqrt_deg_local)
    gy = torch.randn_like(out_c)
    out_c.backward(gy)
    # GPU path
    sh = sh0.to(DEV)
    xg = x_cpu.to(DEV).to(dtype).requires_grad_(True)
    wg = w_cpu.to(DEV).requires_grad_(True)
    out_g = F.scatter_gather_weighted(xg, wg, sh,
                                      dst_scale=sh.rsqrt_deg_local)
    out_g.backward(gy.to(DEV).to(dtype))
    tol = 1e-4 if dtype == torch.float32 else \
        out_c.abs().max().item() * 2 ** -7 + 1e-2
    assert torch.allclose(out_g.float().cpu(), out_c, atol=tol, rtol=0.05)
    tol_w = 1e-3 if dtype == torch.float32 else \
        wc.grad.abs().max().item() * 2 ** -6 + 5e-2
    assert torch.allclose(wg.grad.cpu(), wc.grad, atol=tol_w, rtol=0.05), \
        (wg.grad.cpu() - wc.grad).abs().max()
    tol_x = 1e-4 if dtype == torch.float32 else \
        xc.grad.abs().max().item() * 2 ** -7 + 1e-2
    assert torch.allclose(xg.grad.float().cpu(), xc.grad, atol=tol_x,
                          rtol=0.05)
    # gcn_norm edge weights == the fused normalized path (both on GPU)
    wn = edge_tensor(sh, init="gcn_norm")
    got = F.scatter_gather_weighted(xg.detach(), wn, sh)
    want = F.scatter_gather(xg.detach(), sh, normalize=True)
    assert torch.allclose(got.float(), want.float(), atol=tol, rtol=0.05)


def test_spmm_strips_gpu(monkeypatch):
    """Forced source-strip-blocked scatter_gather (fwd + bwd, fp32
    accumulator + mixed-output kernel) vs the single-pass path."""
    from roc_amd import build_shard
    from roc_amd.graph import synthetic_dataset
    torch.manual_seed(31)
    g, feats, *_ = synthetic_dataset("cora", scale=0.2, seed=5)
    sh_plain = build_shard(g, 0, 1).to(DEV)
    monkeypatch.setenv("ROC_SPMM_STRIP_MIN_EDGES", "0")
    monkeypatch.setenv("ROC_SPMM_STRIP_WIDTH", "97")
    sh_strip = build_shard(g, 0, 1).to(DEV)
    assert sh_strip.fwd_strips is not None
    for dtype in (torch.float32, torch.bfloat16):
        x1 = feats.to(DEV).to(dtype).requires_grad_(True)
        x2 = feats.to(DEV).to(dtype).requires_grad_(True)
        gy = torch.randn(g.num_nodes, feats.shape[1], device=DEV).to(dtype)
        y1 = F.scatter_gather(x1, sh_plain, dst_scale=sh_plain.rsqrt_deg_local)
        y2 = F.scatter_gather(x2, sh_strip, dst_scale=sh_strip.rsqrt_deg_local)
        y1.backward(gy)
        y2.backward(gy)
        tol = 1e-5 if dtype == torch.float32 else \
            y1.float().abs().max().item() * 2 ** -7 + 1e-2
        assert torch.allclose(y1.float(), y2.float(), atol=tol, rtol=0.02), \
            (dtype, (y1.float() - y2.float()).abs().max())
        assert torch.allclose(x1.grad.float(), x2.grad.float(), atol=tol,
                              rtol=0.02), dtype


def test_spmm_col_phases_gpu(monkeypatch):
    """Column-phase strip passes (64-col windows over wide strips) must
    match whole-row strip passes AND the single-pass kernel bit-for-bit
    in fp32 (all three accumulate in source-ascending order)."""
    import roc_amd.ops.functional as Fn
    from roc_amd import build_shard
    from roc_amd.graph import synthetic_graph
    torch.manual_seed(17)
    g = synthetic_graph(1500, 60000, seed=11)
    sh_plain = build_shard(g, 0, 1).to(DEV)
    monkeypatch.setenv("ROC_SPMM_STRIP_MIN_EDGES", "0")
    monkeypatch.setenv("ROC_SPMM_STRIP_WIDTH", "256")
    sh_strip = build_shard(g, 0, 1).to(DEV)
    assert sh_strip.fwd_strips is not None
    for dtype in (torch.float32, torch.bfloat16):
        for D in (128, 192):  # phase applies: D % 64 == 0, >= 2 phases
            x = torch.randn(g.num_nodes, D, device=DEV).to(dtype)
            gy = torch.randn(g.num_nodes, D, device=DEV).to(dtype)
            outs, grads = [], []
            for ph in (0, 64):
                monkeypatch.setattr(Fn, "_PHASE_COLS", ph)
                xg = x.clone().requires_grad_(True)
                y = F.scatter_gather(xg, sh_strip,
                                     dst_scale=sh_strip.rsqrt_deg_local)
                y.backward(gy)
                outs.append(y.float())
                grads.append(xg.grad.float())
            xg = x.clone().requires_grad_(True)
            y = F.scatter_gather(xg, sh_plain,
                                 dst_scale=sh_plain.rsqrt_deg_local)
            y.backward(gy)
            outs.append(y.float())
            grads.append(xg.grad.float())
            monkeypatch.setattr(Fn, "_PHASE_COLS", None)
            for o in outs[1:]:
                assert torch.equal(outs[0], o) if dtype == torch.float32 \
                    else torch.allclose(outs[0], o, atol=1e-2, rtol=0.02)
            for gr in grads[1:]:
                assert torch.equal(grads[0], gr) if dtype == torch.float32 \
                    else torch.allclose(grads[0], gr, atol=1e-2, rtol=0.02)


def test_edge_softmax_gpu():
    """Fused segment-softmax kernel fwd+bwd vs the CPU reference."""
    from roc_amd import build_shard
    from roc_amd.graph import synthetic_dataset
    torch.manual_seed(23)
    g, *_ = synthetic_dataset("cora", scale=0.1, seed=5)
    sh0 = build_shard(g, 0, 1)
    s_cpu = (torch.randn(sh0.num_local_edges) * 3)
    gy = torch.randn(sh0.num_local_edges)
    sc = s_cpu.clone().requires_grad_(True)
    a_c = F.edge_softmax(sc, sh0)
    a_c.backward(gy)
    sh = sh0.to(DEV)
    sg = s_cpu.to(DEV).requires_grad_(True)
    a_g = F.edge_softmax(sg, sh)
    a_g.backward(gy.to(DEV))
    assert torch.allclose(a_g.cpu(), a_c, atol=1e-6), \
        (a_g.cpu() - a_c).abs().max()
    assert torch.allclose(sg.grad.cpu(), sc.grad, atol=1e-6), \
        (sg.grad.cpu() - sc.grad).abs().max()


def test_attention_softmax_gpu():
    """Fused att_softmax kernels (fwd + atomic-scatter bwd) vs CPU."""
    from roc_amd import build_shard
    from roc_amd.graph import synthetic_dataset
    torch.manual_seed(29)
    g, *_ = synthetic_dataset("cora", scale=0.1, seed=5)
    sh0 = build_shard(g, 0, 1)
    src_c = torch.randn(sh0.n_local, requires_grad=True)
    dst_c = torch.randn(sh0.n_local, requires_grad=True)
    gy = torch.randn(sh0.num_local_edges)
    a_c = F.attention_softmax(src_c, dst_c, sh0)
    a_c.backward(gy)
    sh = sh0.to(DEV)
    src_g = src_c.detach().to(DEV).requires_grad_(True)
    dst_g = dst_c.detach().to(DEV).requires_grad_(True)
    a_g = F.attention_softmax(src_g, dst_g, sh)
    a_g.backward(gy.to(DEV))
    assert torch.allclose(a_g.cpu(), a_c, atol=1e-6)
    assert torch.allclose(src_g.grad.cpu(), src_c.grad, atol=1e-4), \
        (src_g.grad.cpu() - src_c.grad).abs().max()
    assert torch.allclose(dst_g.grad.cpu(), dst_c.grad, atol=1e-5)


def test_gat_step_gpu():
    """One GAT train epoch on GPU (bf16): finite metrics, attention
    grads flow through edge_softmax + spmm_edge kernels."""
    from roc_amd import build_shard, build_model, AdamOptimizer, Trainer
    from roc_amd.graph import synthetic_dataset
    torch.manual_seed(0)
    g, feats, labels, mask, c = synthetic_dataset("cora", scale=0.2, seed=3)
    sh = build_shard(g, 0, 1)
    model = build_model("gat", [feats.shape[1], 32, c], dropout=0.2,
                        seed=1, heads=4)
    opt = AdamOptimizer(model.parameters(), lr=0.01, weight_decay=1e-4)
    tr = Trainer(model, sh, feats, labels, mask, opt, device=DEV,
                 compute_dtype=torch.bfloat16)
    m0 = tr.evaluate()
    for _ in range(5):
        tr.train_epoch()
    m1 = tr.evaluate()
    assert m1["ce_loss"] == m1["ce_loss"]  # finite
    assert m1["ce_loss"] < m0["ce_loss"] + 0.1
    assert any(p.grad is not None and p.grad.abs().sum() > 0
               for p in model.a_src)


@pytest.mark.parametrize("c", [107, 172])
def test_softmax_ce_wide_unpadded_gpu(c):
    """C > 64 takes the wide path (softmax_ce.hip multi-pass); run it
    UNPADDED at the amazon (107) and papers (172) class counts against
    the fp32 reference — fwd metrics AND backward grad."""
    torch.manual_seed(21)
    n = 1500
    logits = torch.randn(n, c, device=DEV, requires_grad=True)
    labels = torch.randint(0, c, (n,), device=DEV)
    mask = torch.randint(1, 4, (n,), dtype=torch.int32, device=DEV)
    loss, metrics = F.softmax_cross_entropy(logits, labels, mask)
    loss.backward()
    md = F.decode_metrics(metrics)
    lc = logits.detach().cpu().requires_grad_(True)
    dl_want, md_want = ref.softmax_cross_entropy(
        lc.detach().float(), labels.cpu(), mask.cpu())
    assert md["train_total"] == md_want["train_total"]
    assert md["train_acc"] == pytest.approx(
        md_want["train_correct"] / md_want["train_total"], abs=1e-3)
    assert md["ce_loss"] == pytest.approx(md_want["ce_loss"], rel=3e-2)
    assert torch.allclose(logits.grad.cpu(), dl_want, atol=1e-4), \
        (logits.grad.cpu() - dl_want).abs().max()


def test_softmax_ce_grad_gpu():
    torch.manual_seed(10)
    n, c = 500, 41
    logits = torch.randn(n, c, device=DEV, requires_grad=True)
    labels = torch.randint(0, c, (n,), device=DEV)
    mask = torch.randint(1, 4, (n,), dtype=torch.int32, device=DEV)
    loss, _ = F.softmax_cross_entropy(logits, labels, mask)
    loss.backward()
    lc = logits.detach().cpu().requires_grad_(True)
    loss_c, _ = F.softmax_cross_entropy(lc, labels.cpu(), mask.cpu())
    loss_c.backward()
    assert torch.allclose(logits.grad.cpu(), lc.grad, atol=1e-4)


def test_adam_gpu():
    torch.manual_seed(11)
    n = 12345
    w = torch.randn(n)
    g = torch.randn(n)
    m = torch.randn(n).abs()
    v = torch.randn(n).abs()
    wd, md_, vd = w.to(DEV), m.to(DEV), v.to(DEV)
    F.adam_step(wd, g.to(DEV), md_, vd, 0.01, 0.9, 0.999, 1e-8, 1e-4)
    ref.adam_step(w, g, m, v, 0.01, 0.9, 0.999, 1e-8, 1e-4)
    assert torch.allclose(wd.cpu(), w, atol=1e-6)
    assert torch.allclose(md_.cpu(), m, atol=1e-6)
    assert torch.allclose(vd.cpu(), v, atol=1e-6)


def test_spmm_accumulate_mode(graph_small):
    """Two-pass aggregation (edge split + accumulate flag, the halo-overlap
    building block) must equal the single-pass result."""
    g, shard = graph_small
    torch.manual_seed(14)
    n = g.num_nodes
    x = torch.randn(n, 64).to(torch.bfloat16).to(DEV)
    rowptr = g.rowptr.to(DEV)
    colidx = g.colidx.to(DEV)
    deg = (g.rowptr[1:] - g.rowptr[:-1]).float().clamp(min=1)
    dst = deg.rsqrt().to(DEV)
    want = torch.empty_like(x)
    _ext().spmm(want, x, rowptr, colidx, dst, None, None)
    # split each row's edges at the midpoint into two CSRs
    rp = g.rowptr
    mid = (rp[:-1] + (rp[1:] - rp[:-1]) // 2)
    rp1 = torch.zeros(n + 1, dtype=torch.int64)
    rp1[1:] = torch.cumsum(mid - rp[:-1], 0)
    rp2 = torch.zeros(n + 1, dtype=torch.int64)
    rp2[1:] = torch.cumsum(rp[1:] - mid, 0)
    ci1 = torch.cat([g.colidx[rp[v]:mid[v]] for v in range(n)])
    ci2 = torch.cat([g.colidx[mid[v]:rp[v + 1]] for v in range(n)])
    got = torch.empty_like(x)
    _ext().spmm(got, x, rp1.to(DEV), ci1.to(DEV), None, None, None, False)
    _ext().spmm(got, x, rp2.to(DEV), ci2.to(DEV), dst, None, None, True)
    # pass-2 starts from bf16 partials: one extra rounding step
    tol = want.float().abs().max().item() * 2 ** -6
    assert torch.allclose(got.float(), want.float(), atol=tol, rtol=0.05), \
        (got.float() - want.float()).abs().max()


def test_empty_and_odd_shapes():
    """Degenerate partitions (0 rows) and odd dims must not fault."""
    A0 = torch.empty(0, 32, dtype=torch.bfloat16, device=DEV)
    B = torch.randn(16, 32, dtype=torch.bfloat16, device=DEV)
    C0 = torch.empty(0, 16, dtype=torch.bfloat16, device=DEV)
    _ext().gemm_rr(C0, A0, B, False)
    dw = torch.zeros(32, 8, dtype=torch.float32, device=DEV)
    _ext().gemm_atb(dw, A0, torch.empty(0, 8, dtype=torch.bfloat16,
                                        device=DEV))
    assert dw.abs().sum().item() == 0.0
    torch.cuda.synchronize()


def test_spmm_shape_sweep(graph_small):
    """Kernel dispatch sweep: every (dtype, team-size, tail) combination
    vs the torch reference on a real graph."""
    g, shard = graph_small
    shd = shard.to(DEV)
    for D in (8, 16, 24, 48, 63, 64, 65, 120, 128, 200, 256, 320, 513):
        for dt in (torch.float32, torch.bfloat16):
            x = torch.randn(g.num_nodes, D).to(dt)
            want = ref.spmm(x.float(), g.rowptr, g.colidx, g.num_nodes)
            got = F.scatter_gather(x.to(DEV), shd).float().cpu()
            tol = want.abs().max().item() * (2 ** -7 if dt == torch.bfloat16
                                             else 1e-5) + 1e-3
            assert torch.allclose(got, want, atol=tol, rtol=0.05), \
                (D, dt, (got - want).abs().max())


def test_ag_split_spmm_blocks_match_reference():
    """The allgather-overlap kernel shapes: self/remote CSR splits with
    degree-desc row orders, and per-owner-block transpose slices with
    block-local orders + accumulate — exactly what the multi-GPU
    aggregate path launches (parallel/aggregate.py)."""
    from roc_amd.parallel.halo import _spmm_part
    from roc_amd.parallel.aggregate import _ag_block_bounds
    g = synthetic_graph(800, 24000, seed=21)
    sh = build_shard(g, 0, 2)          # rank 0's view of a 2-way split
    assert sh.comm_mode == "allgather" and sh.ag_self_rowptr is not None
    D, ws, mr = 64, sh.world_size, sh.ag_max_rows
    torch.manual_seed(3)
    x_local = torch.randn(sh.n_local, D)
    x_full = torch.randn(g.num_nodes, D)
    x_full[:sh.n_local] = x_local
    # padded gather-space buffer [ws*mr, D]
    gathered = torch.zeros(ws * mr, D)
    for r in range(ws):
        blo, bhi = sh.bounds[r], sh.bounds[r + 1]
        gathered[r * mr:r * mr + (bhi - blo)] = x_full[blo:bhi]
    shd = sh.to(DEV)
    # forward: self pass (order) + remote accumulate (order)
    out = torch.empty(sh.n_local, D, device=DEV)
    _spmm_part(out, x_local.to(DEV), shd.ag_self_rowptr, shd.ag_self_colidx,
               None, False, shd.ag_self_row_order)
    _spmm_part(out, gathered.to(DEV), shd.ag_rem_rowptr, shd.ag_rem_colidx,
               None, True, shd.ag_rem_row_order)
    want = ref.spmm(x_full, g.rowptr, g.colidx, g.num_nodes)[:sh.n_local]
    assert torch.allclose(out.cpu(), want, rtol=1e-4, atol=1e-3), \
        (out.cpu() - want).abs().max()
    # backward blocks: per-owner slice + block-local order
    dy = torch.randn(sh.n_local, D)
    bnds = _ag_block_bounds(shd)
    dfull = torch.empty(ws * mr, D, device=DEV)
    dyd = dy.to(DEV)
    for r in range(ws):
        rp_blk = shd.ag_t_rowptr[r * mr:(r + 1) * mr + 1]
        rp_blk = (rp_blk - rp_blk[:1]).contiguous()
        cols = shd.ag_t_colidx[bnds[r]:bnds[r + 1]]
        order = shd.ag_t_blk_order[r * mr:(r + 1) * mr]
        _spmm_part(dfull[r * mr:(r + 1) * mr], dyd, rp_blk, cols,
                   None, False, order)
    want_full = ref.spmm(dy, sh.ag_t_rowptr, sh.ag_t_colidx, ws * mr)
    assert torch.allclose(dfull.cpu(), want_full, rtol=1e-4, atol=1e-3), \
        (dfull.cpu() - want_full).abs().max()


def test_spmm_schedule_equivalence():
    # the row schedule (natural vs degree-desc vs any permutation) is a
    # pure performance knob: each row is reduced by one team in the same
    # in-row order, so outputs must be BIT-identical
    from roc_amd import _C
    g = synthetic_graph(40000, 800_000, seed=13, locality=0.7,
                        num_communities=10)
    D = 128
    rowptr = g.rowptr.to(DEV)
    colidx = g.colidx.to(DEV)
    deg = (g.rowptr[1:] - g.rowptr[:-1]).float().clamp(min=1)
    rsq = deg.rsqrt().to(DEV)
    x = torch.randn(g.num_nodes, D).to(torch.bfloat16).to(DEV)
    out_nat = torch.empty_like(x)
    out_deg = torch.empty_like(x)
    _C.spmm(out_nat, x, rowptr, colidx, rsq, None, None)
    order = torch.argsort(-deg).int().to(DEV)
    _C.spmm(out_deg, x, rowptr, colidx, rsq, None, order)
    assert torch.equal(out_nat, out_deg)
