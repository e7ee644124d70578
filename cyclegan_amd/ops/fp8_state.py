"""Delayed per-tensor scaling state for the fp8 forward path.

Transformer-engine style: each fp8 conv layer (keyed by its master
weight) owns a slot pair {amax_prev, amax_cur} in ONE device arena.
quant_fp8_d quantizes this step's activations with the PREVIOUS step's
amax (no separate full-tensor reduction) while block-reducing this
step's max|x| into amax_cur; the trainer rolls prev <- cur once per
step (one kernel over the whole arena — hip-graph capturable because
every address is fixed at first use)."""

from __future__ import annotations

import torch
from torch.utils.weak import WeakTensorKeyDictionary

from . import backend

CAP = 512

_arena = None
_slots: "WeakTensorKeyDictionary" = WeakTensorKeyDictionary()
_used = 0


def slots_for(w: torch.Tensor):
    """(amax_prev view, amax_cur view, fresh) for layer keyed by w."""
    global _arena, _used
    if _arena is None or _arena.device != w.device:
        _arena = torch.zeros(2, CAP, dtype=torch.float32, device=w.device)
    ent = _slots.get(w)
    fresh = ent is None
    if fresh:
        assert _used < CAP, "fp8 amax arena exhausted"
        i = _used
        _used += 1
        ent = (_arena[0, i:i + 1], _arena[1, i:i + 1])
        _slots[w] = ent
    return ent[0], ent[1], fresh


def roll():
    """prev <- cur, cur <- 0 for every registered layer (call once per
    optimizer step while fp8 mode is on)."""
    if _arena is not None and _used:
        backend.ext().amax_roll(_arena, _used)


def reset():
    global _arena, _used
    _arena = None
    _slots.clear()
    _used = 0
