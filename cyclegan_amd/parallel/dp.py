"""Synchronous data-parallel engine: one process per GPU, RCCL over xGMI.

Replaces tf.distribute.MirroredStrategy (reference main.py:370). Semantics
preserved exactly:
- losses are pre-scaled by 1/global_batch_size (main.py:172-174), so the
  gradient all-reduce is a plain SUM (S1-S4 in SURVEY §2.3);
- four per-optimizer gradient groups, all-reduced as four flat buffers;
  each group's all-reduce is issued async right after that group's
  backward, overlapping with the next group's backward (the reference runs
  the 4 minimize() calls serially — main.py:249-260 — we hide the ~113 MB
  of per-step gradient traffic behind compute instead);
- per-step scalar metrics are summed on-device and all-reduced once per
  epoch (mathematically identical to the reference's per-step S5 reduce
  followed by an epoch mean, without a per-step host sync).

Backend: "nccl" (RCCL on ROCm) on GPU, "gloo" on CPU (CI).
"""

from __future__ import annotations

import os
from typing import Optional

import torch
import torch.distributed as dist


class DistContext:
    def __init__(self, device: Optional[torch.device] = None):
        self.rank = int(os.environ.get("RANK", "0"))
        self.world_size = int(os.environ.get("WORLD_SIZE", "1"))
        self.local_rank = int(os.environ.get("LOCAL_RANK", str(self.rank)))
        # CYG_FORCE_DIST=1: create the RCCL communicator even at
        # world_size 1 (on-silicon shakeout of the ProcessGroupNCCL path
        # within a 1-GPU lease; RCCL refuses two ranks on one device —
        # "Duplicate GPU detected", see profiles/rccl_shakeout.md)
        self.distributed = (self.world_size > 1 or
                            os.environ.get("CYG_FORCE_DIST") == "1")
        if device is not None:
            self.device = device
        elif torch.cuda.is_available():
            # modulo: lets an N-rank job oversubscribe fewer GPUs (RCCL
            # shakeout of the multi-rank path on a 1-GPU box)
            self.device = torch.device(
                "cuda", self.local_rank % torch.cuda.device_count())
        else:
            self.device = torch.device("cpu")
        if self.device.type == "cuda":
            torch.cuda.set_device(self.device)
        if self.distributed and not dist.is_initialized():
            backend = "nccl" if self.device.type == "cuda" else "gloo"
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29500")
            dist.init_process_group(backend=backend, rank=self.rank,
                                    world_size=self.world_size)

    @property
    def is_main(self) -> bool:
        return self.rank == 0

    def barrier(self):
        if self.distributed:
            dist.barrier()

    def broadcast_module(self, module: torch.nn.Module):
        """Mirror rank-0 initial weights to all replicas (reference S7)."""
        if not self.distributed:
            return
        for t in module.state_dict().values():
            if isinstance(t, torch.Tensor):
                dist.broadcast(t, src=0)

    def broadcast_tensor(self, t: torch.Tensor):
        if self.distributed:
            dist.broadcast(t, src=0)

    def all_reduce_(self, t: torch.Tensor):
        """Blocking SUM all-reduce in place."""
        if self.distributed:
            dist.all_reduce(t, op=dist.ReduceOp.SUM)

    def all_reduce_async(self, t: torch.Tensor):
        """Async SUM all-reduce; returns a handle with .wait() (or None)."""
        if not self.distributed:
            return None
        return dist.all_reduce(t, op=dist.ReduceOp.SUM, async_op=True)


class GradSync:
    """Issues one async flat all-reduce per model group, waited before the
    optimizer steps.

    With ``timing=True`` (CUDA only) each all-reduce is bracketed by hip
    events on the comm stream; ``pop_comm_ms()`` returns the accumulated
    RCCL time since the last read (one host sync per read — call it at
    epoch granularity, not per step)."""

    def __init__(self, ctx: DistContext, timing: bool = False):
        self.ctx = ctx
        self._pending = []
        self._timer = None
        if timing and ctx.device.type == "cuda":
            from ..utils.profiler import HipEventTimer
            self._timer = HipEventTimer()

    def launch(self, flat_grad: torch.Tensor):
        if self._timer is not None and self.ctx.distributed:
            self._timer.begin()
        h = self.ctx.all_reduce_async(flat_grad)
        if h is not None:
            self._pending.append(h)
        if self._timer is not None and self.ctx.distributed:
            self._timer.end()

    def wait_all(self):
        for h in self._pending:
            h.wait()
        self._pending.clear()

    def pop_comm_ms(self) -> float:
        """Accumulated all-reduce GPU time (ms) since last call; 0 if
        timing is off or single-rank."""
        return self._timer.elapsed_ms() if self._timer is not None else 0.0
