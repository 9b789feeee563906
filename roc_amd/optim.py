"""Adam optimizer with L2-coupled weight decay and step-decay LR schedule.

Semantics match the reference (`optimizer_kernel.cu:43-63`,
`optimizer.cc:79-85`, `gnn.cc:99-101`):
  gt = g + wd * w;   m, v EMA;   w -= alpha_t * m / (sqrt(v) + eps)
  alpha_t = lr_t * sqrt(1 - beta2^t) / (1 - beta1^t)
  lr_t = lr * decay_rate ** (t // decay_steps)
The per-parameter update is one fused CDNA4 kernel on GPU.
"""
from __future__ import annotations

import math
from typing import Iterable

import torch

from .ops import functional as F


class AdamOptimizer:
    def __init__(self, params: Iterable[torch.nn.Parameter], lr: float = 0.01,
                 weight_decay: float = 1e-4, beta1: float = 0.9,
                 beta2: float = 0.999, eps: float = 1e-8,
                 decay_rate: float = 1.0, decay_steps: int = 100):
        self.params = [p for p in params if p.numel() > 0]
        self.lr = lr
        self.weight_decay = weight_decay
        self.beta1, self.beta2, self.eps = beta1, beta2, eps
        self.decay_rate, self.decay_steps = decay_rate, decay_steps
        self.t = 0
        self.m = [torch.zeros_like(p.data) for p in self.params]
        self.v = [torch.zeros_like(p.data) for p in self.params]
        self._step_dev = None  # device schedule (hipGraph capture mode)
        self._flat_grad = None

    def setup_flat_grads(self) -> torch.Tensor:
        """Point every param's .grad into ONE flat fp32 buffer: autograd
        accumulates in place, the gradient all-reduce is a single
        collective on the buffer (no cat/copy-back), and zero_grad is one
        fill. Call after the model is on its final device."""
        total = sum(p.numel() for p in self.params)
        dev = self.params[0].device if self.params else "cpu"
        self._flat_grad = torch.zeros(total, dtype=torch.float32, device=dev)
        off = 0
        for p in self.params:
            n = p.numel()
            p.grad = self._flat_grad[off:off + n].view_as(p)
            off += n
        return self._flat_grad

    def set_device_step(self, step_tensor) -> None:
        """int64[1] device counter (bumped by the engine inside the
        captured epoch); the Adam kernel derives alpha_t from it."""
        self._step_dev = step_tensor

    def zero_grad(self) -> None:
        if self._flat_grad is not None:
            self._flat_grad.zero_()
            return
        for p in self.params:
            if p.grad is not None:
                p.grad.detach_()
                p.grad.zero_()

    def current_lr(self) -> float:
        return self.lr * self.decay_rate ** (self.t // max(self.decay_steps, 1))

    def step(self) -> None:
        self.t += 1
        if self._step_dev is None:
            lr_t = self.current_lr()
            alpha_t = lr_t * math.sqrt(1.0 - self.beta2 ** self.t) / (
                1.0 - self.beta1 ** self.t)
            step_kw = {}
        else:
            alpha_t = self.lr  # device derives decayed, bias-corrected rate
            step_kw = {"step": self._step_dev,
                       "decay_rate": self.decay_rate,
                       "decay_steps": self.decay_steps}
        for i, p in enumerate(self.params):
            if p.grad is None:
                continue
            if self.m[i].device != p.device:  # model moved after ctor
                self.m[i] = self.m[i].to(p.device)
                self.v[i] = self.v[i].to(p.device)
            F.adam_step(p.data, p.grad.data, self.m[i], self.v[i], alpha_t,
                        self.beta1, self.beta2, self.eps, self.weight_decay,
                        **step_kw)
        F.bump_weight_version()  # invalidate cached weight casts

    def state_dict(self) -> dict:
        return {"t": self.t, "m": self.m, "v": self.v,
                "lr": self.lr, "weight_decay": self.weight_decay,
                "beta1": self.beta1, "beta2": self.beta2, "eps": self.eps,
                "decay_rate": self.decay_rate, "decay_steps": self.decay_steps}

    def load_state_dict(self, sd: dict) -> None:
        self.t = sd["t"]
        for dst, src in zip(self.m, sd["m"]):
            dst.copy_(src)
        for dst, src in zip(self.v, sd["v"]):
            dst.copy_(src)
