"""Fused Adam over flat fp32 parameter/grad buffers.

Replicates the Keras/TF2 Adam update used by the reference exactly
(/root/reference/main.py:134-145: lr=2e-4, beta1=0.5, beta2=0.9, eps=1e-7):

    lr_t = lr * sqrt(1 - b2^t) / (1 - b1^t)
    m    = b1*m + (1-b1)*g
    v    = b2*v + (1-b2)*g^2
    p   -= lr_t * m / (sqrt(v) + eps)

Note the epsilon sits next to the *uncorrected* sqrt(v) (TF formula) —
PyTorch's Adam applies eps to sqrt(v_hat); with eps=1e-7 the difference is
tiny but we match TF for parity.

On MI355X this is ONE kernel per model over the flat buffers (the whole
generator is a single 11.4M-element update — zero per-tensor launch
overhead), which also refreshes nothing else: bf16 shadow weights recast
lazily via the version counter (ops.shadow).
"""

from __future__ import annotations

import math

import torch

from . import backend


class FusedAdam:
    """Adam on a (flat_param, flat_grad) fp32 pair."""

    def __init__(self, flat_param: torch.Tensor, flat_grad: torch.Tensor,
                 lr: float = 2e-4, beta1: float = 0.5, beta2: float = 0.9,
                 eps: float = 1e-7):
        assert flat_param.dtype == torch.float32
        self.p = flat_param
        self.g = flat_grad
        self.lr, self.b1, self.b2, self.eps = lr, beta1, beta2, eps
        self.m = torch.zeros_like(flat_param)
        self.v = torch.zeros_like(flat_param)
        self.t = 0
        # graph-capture mode: the update kernel reads lr_t from this device
        # scalar instead of an immediate, so a captured step follows the
        # per-step bias correction written by advance_lr() before replay
        self.graph_mode = False
        self._lr_t_dev = None

    def _lr_t(self) -> float:
        return (self.lr * math.sqrt(1 - self.b2 ** self.t)
                / (1 - self.b1 ** self.t))

    @torch.no_grad()
    def prepare_graph(self):
        """Switch to the capture-safe kernel and allocate the device lr_t
        scalar (value is a placeholder: capture records, never executes)."""
        self.graph_mode = True
        if self._lr_t_dev is None:
            self._lr_t_dev = torch.zeros(1, dtype=torch.float32,
                                         device=self.p.device)

    @torch.no_grad()
    def advance_lr(self):
        """t += 1 and refresh the device lr_t scalar (graph mode: called
        once per step OUTSIDE the captured region)."""
        self.t += 1
        if self._lr_t_dev is None:
            self._lr_t_dev = torch.empty(1, dtype=torch.float32,
                                         device=self.p.device)
        self._lr_t_dev.fill_(self._lr_t())

    @torch.no_grad()
    def step(self):
        if self.graph_mode:
            # t/lr_t advanced by advance_lr(); capture-safe kernel
            backend.ext().adam_step_dev(self.p, self.g, self.m, self.v,
                                        self._lr_t_dev, self.b1, self.b2,
                                        self.eps)
            return
        self.t += 1
        if backend.use_hip(self.p):
            backend.ext().adam_step(self.p, self.g, self.m, self.v,
                                    self.lr, self.b1, self.b2, self.eps, self.t)
            # the raw kernel bypasses the dispatcher, so the in-place update
            # does not bump the version counter that invalidates the bf16
            # shadow-weight caches (ops.shadow) — bump it explicitly, or
            # compute would keep running on the step-0 weights forever
            torch.autograd.graph.increment_version(self.p)
            return
        lr_t = self.lr * math.sqrt(1 - self.b2 ** self.t) / (1 - self.b1 ** self.t)
        self.m.mul_(self.b1).add_(self.g, alpha=1 - self.b1)
        self.v.mul_(self.b2).addcmul_(self.g, self.g, value=1 - self.b2)
        self.p.addcdiv_(self.m, self.v.sqrt().add_(self.eps), value=-lr_t)

    def state_dict(self):
        return {"m": self.m, "v": self.v, "t": self.t,
                "lr": self.lr, "b1": self.b1, "b2": self.b2, "eps": self.eps}

    def load_state_dict(self, sd):
        self.m.copy_(sd["m"])
        self.v.copy_(sd["v"])
        self.t = sd["t"]
        self.lr, self.b1, self.b2, self.eps = sd["lr"], sd["b1"], sd["b2"], sd["eps"]
