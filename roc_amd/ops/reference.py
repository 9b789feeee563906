"""Plain-PyTorch reference implementations of every operator.

These are the numerics oracle for the HIP kernels (tests compare the
CDNA4 kernels against these in fp32) and the execution path for the
CPU-only Cora config. Semantics mirror the reference ops
(`/root/reference/*_kernel.cu`), cited per function.
"""
from __future__ import annotations

import torch

from ..graph import MASK_TRAIN, MASK_VAL, MASK_TEST


def spmm(x: torch.Tensor, rowptr: torch.Tensor, colidx: torch.Tensor,
         num_rows: int) -> torch.Tensor:
    """out[v] = sum_{u in N_in(v)} x[u]  (reference `scattergather_kernel.cu:20-76`).

    x may have more rows than num_rows (halo-extended input).
    """
    deg = (rowptr[1:] - rowptr[:-1]).to(torch.long)
    dst = torch.repeat_interleave(
        torch.arange(num_rows, dtype=torch.long, device=x.device), deg
    )
    out = torch.zeros(num_rows, x.shape[1], dtype=x.dtype, device=x.device)
    out.index_add_(0, dst, x[colidx.to(torch.long)])
    return out


def spmm_weighted(x: torch.Tensor, rowptr: torch.Tensor,
                  colidx: torch.Tensor, edge_val: torch.Tensor,
                  num_rows: int) -> torch.Tensor:
    """out[v] = sum_{e in row v} edge_val[e] * x[col_e]  — the edge-tensor
    consumer op (the reference declares per-edge tensors,
    `gnn.cc:475-623` create_edge_tensor, but ships no op over them)."""
    deg = (rowptr[1:] - rowptr[:-1]).to(torch.long)
    dst = torch.repeat_interleave(
        torch.arange(num_rows, dtype=torch.long, device=x.device), deg
    )
    out = torch.zeros(num_rows, x.shape[1], dtype=x.dtype, device=x.device)
    out.index_add_(0, dst,
                   x[colidx.to(torch.long)] * edge_val.unsqueeze(1).to(x.dtype))
    return out


def edge_dot(dy: torch.Tensor, x: torch.Tensor, rowptr: torch.Tensor,
             colidx: torch.Tensor) -> torch.Tensor:
    """dval[e] = <dy[row_e], x[col_e]> — gradient of spmm_weighted wrt
    the per-edge values (fp32 accumulate)."""
    num_rows = rowptr.numel() - 1
    deg = (rowptr[1:] - rowptr[:-1]).to(torch.long)
    dst = torch.repeat_interleave(
        torch.arange(num_rows, dtype=torch.long, device=dy.device), deg
    )
    return (dy[dst].float() * x[colidx.to(torch.long)].float()).sum(dim=1)


def degnorm(x: torch.Tensor, deg: torch.Tensor) -> torch.Tensor:
    """out[v] = x[v] / sqrt(indeg(v))  (reference `graphnorm_kernel.cu:19-57`)."""
    return x * torch.rsqrt(deg.clamp(min=1.0)).unsqueeze(1).to(x.dtype)


def linear(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """y = x @ w, w stored [in_dim, out_dim] (reference `linear_kernel.cu:76-80`)."""
    return x @ w.to(x.dtype)


def relu(x: torch.Tensor) -> torch.Tensor:
    return torch.relu(x)


def relu_grad(dy: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
    """dx = dy where y > 0 (reference `linear_kernel.cu:120-127` masks on output)."""
    return dy * (y > 0).to(dy.dtype)


def sigmoid(x: torch.Tensor) -> torch.Tensor:
    return torch.sigmoid(x)


def sigmoid_grad(dy: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
    return dy * y * (1.0 - y)


def dropout_mask(shape, p: float, seed: int, offset: int, device) -> torch.Tensor:
    """Deterministic dropout keep-mask from (seed, offset).

    CPU reference for the Philox kernel: uses torch's generator; the HIP
    kernel regenerates its own Philox stream, so GPU-vs-CPU tests compare
    statistics / identity-at-infer, not the exact mask.
    """
    gen = torch.Generator(device="cpu").manual_seed(seed * 0x9E3779B1 + offset)
    return (torch.rand(shape, generator=gen) >= p).to(device)


def softmax_cross_entropy(
    logits: torch.Tensor,
    labels: torch.Tensor,
    mask: torch.Tensor,
    grad_scale: float = 1.0,
    num_classes=None,
):
    """Fused softmax -> CE gradient with train-mask zeroing + metrics.

    Reference `softmax_kernel.cu:19-79`:
      dlogit = softmax(logit) - onehot(label), zeroed where mask != Train
      roc_loss = sum over train rows of (1 - p_true)   (the reference's "loss")
    We additionally report true mean cross-entropy over train rows.
    num_classes < logits width: softmax over the first num_classes cols
    only (padded class dim); pad columns get zero gradient.
    Returns (dlogits, metrics dict).
    """
    stride = logits.shape[1]
    C = num_classes or stride
    if C < stride:
        dl_full = torch.zeros_like(logits)
        dl, md = softmax_cross_entropy(
            logits[:, :C].contiguous(), labels, mask, grad_scale)
        dl_full[:, :C] = dl
        return dl_full, md
    # fp32 accumulation for low-precision inputs (the HIP kernel's
    # behavior); fp32/fp64 inputs keep their own precision (fp64 matters
    # for gradcheck)
    lf = logits if logits.dtype in (torch.float32, torch.float64) \
        else logits.to(torch.float32)
    p = torch.softmax(lf, dim=1)
    n, c = lf.shape
    onehot = torch.zeros_like(p)
    onehot[torch.arange(n, device=lf.device), labels] = 1.0
    train = (mask == MASK_TRAIN)
    dl = (p - onehot) * train.unsqueeze(1).to(p.dtype) * grad_scale
    p_true = p[torch.arange(n, device=lf.device), labels]
    pred = p.argmax(dim=1)
    correct = (pred == labels)
    metrics = {}
    metrics["roc_loss"] = float((1.0 - p_true)[train].sum())
    metrics["ce_loss"] = float((-torch.log(p_true.clamp(min=1e-12))[train]).mean()) if train.any() else 0.0
    for name, m in (("train", MASK_TRAIN), ("val", MASK_VAL), ("test", MASK_TEST)):
        sel = (mask == m)
        metrics[f"{name}_total"] = int(sel.sum())
        metrics[f"{name}_correct"] = int(correct[sel].sum())
    return dl.to(logits.dtype), metrics


def adam_step(
    w: torch.Tensor,
    g: torch.Tensor,
    m: torch.Tensor,
    v: torch.Tensor,
    alpha_t: float,
    beta1: float,
    beta2: float,
    eps: float,
    weight_decay: float,
) -> None:
    """Fused Adam with L2-coupled decay (reference `optimizer_kernel.cu:43-63`):
      gt = g + wd * w;  m,v EMA;  w -= alpha_t * m / (sqrt(v) + eps)
    alpha_t is the bias-corrected step size (reference `optimizer.cc:79-85`).
    In-place on w, m, v.
    """
    gt = g.to(torch.float32) + weight_decay * w
    m.mul_(beta1).add_(gt, alpha=1.0 - beta1)
    v.mul_(beta2).addcmul_(gt, gt, value=1.0 - beta2)
    w.sub_(alpha_t * m / (v.sqrt() + eps))


def glorot_uniform(shape, seed: int) -> torch.Tensor:
    """GlorotUniform: U[-s, s], s = sqrt(6/(in+out))
    (reference `initializer_kernel.cu:22-51`)."""
    fan_in, fan_out = shape[0], shape[1]
    s = (6.0 / (fan_in + fan_out)) ** 0.5
    gen = torch.Generator().manual_seed(seed)
    return (torch.rand(shape, generator=gen, dtype=torch.float32) * 2.0 - 1.0) * s
