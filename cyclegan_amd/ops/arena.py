"""Shadow arena: all bf16 compute-weight forms of a model group in ONE
flat buffer, refreshed by ONE gather kernel per optimizer step.

Masters live in a group's flat fp32 buffer (parallel.flat.FlatParamGroup).
Every conv needs derived bf16 forms — padded OHWI for forward, padded
channel-transposed [I,kh,kw,O] for dgrad, padded bias — which round 1
rebuilt per master via ~180 per-step ATen pad/cast/permute launches
(profiles/kernel_stats_final.csv: ~9% of GPU time in Fill/copy kernels).
Here each form is a view into one bf16 arena filled by shadow_gather
(shadow.hip) from a precomputed int32 index map; -1 entries produce the
alignment-pad zeros. The refresh is captured inside the hip-graph step.

GPU-only: CPU paths keep the version-keyed caches in ops.shadow (fp32
compute needs no casting there).
"""

from __future__ import annotations

from typing import Dict, List, Tuple

import torch

from . import backend

_PAD = 8  # glds kernels need channel dims >= 8


def _conv_like(m: torch.nn.Module) -> bool:
    return type(m).__name__ == "ConvNHWC"


def _convt_like(m: torch.nn.Module) -> bool:
    return type(m).__name__ == "ConvTransposeNHWC"


def _idx_p(base: int, shape) -> torch.Tensor:
    """Padded OHWI forward form: [O',KH,KW,I'], O'=max(O,8), I'=max(I,8)."""
    O, KH, KW, I = shape
    Op, Ip = max(O, _PAD), max(I, _PAD)
    o = torch.arange(Op).view(-1, 1, 1, 1)
    kh = torch.arange(KH).view(1, -1, 1, 1)
    kw = torch.arange(KW).view(1, 1, -1, 1)
    i = torch.arange(Ip).view(1, 1, 1, -1)
    idx = base + ((o * KH + kh) * KW + kw) * I + i
    idx = torch.where((o < O) & (i < I), idx, torch.tensor(-1))
    return idx.to(torch.int32)


def _idx_tp(base: int, shape) -> torch.Tensor:
    """Padded channel-transposed dgrad form: [I',KH,KW,O']."""
    return _idx_p(base, shape).permute(3, 1, 2, 0).contiguous()


def _idx_plain(base: int, shape) -> torch.Tensor:
    O, KH, KW, I = shape
    return (base + torch.arange(O * KH * KW * I)).view(O, KH, KW, I).to(torch.int32)


def _idx_t(base: int, shape) -> torch.Tensor:
    return _idx_plain(base, shape).permute(3, 1, 2, 0).contiguous()


def _idx_bias(base: int, n: int) -> torch.Tensor:
    np_ = max(n, _PAD)
    idx = base + torch.arange(np_)
    idx[n:] = -1
    return idx.to(torch.int32)


def head_packable(shape, stride, pad_mode, padding) -> bool:
    """Generator-head geometry: 7x7 stride-1 reflect(3) conv with tiny
    Cout and Cin divisible by 8 — runs as the 8-pixel-packed conv (see
    ops.conv._ConvHeadPackedFn)."""
    O, KH, KW, I = shape
    return (O <= 8 and KH == 7 and KW == 7 and stride == 1 and I % 8 == 0
            and I >= 8 and pad_mode == "reflect"
            and tuple(padding) == (3, 3, 3, 3))


def _idx_packp(base: int, shape) -> torch.Tensor:
    """Packed head weight [8*8, KH, 2, 8*I]: n = d*8 + co packs 8 adjacent
    output pixels into channels; k covers a 2-block (16-col) window; slot
    (d, co, ty, blk, pos, ci) maps to w[co][ty][blk*8+pos-d][ci] when the
    tap is in range, else zero (idx -1)."""
    O, KH, KW, I = shape
    assert KW == 7
    d = torch.arange(8).view(-1, 1, 1, 1, 1, 1)
    co = torch.arange(8).view(1, -1, 1, 1, 1, 1)
    ty = torch.arange(KH).view(1, 1, -1, 1, 1, 1)
    blk = torch.arange(2).view(1, 1, 1, -1, 1, 1)
    pos = torch.arange(8).view(1, 1, 1, 1, -1, 1)
    ci = torch.arange(I).view(1, 1, 1, 1, 1, -1)
    tx = blk * 8 + pos - d
    idx = base + ((co * KH + ty) * KW + tx) * I + ci
    ok = (tx >= 0) & (tx < KW) & (co < O)
    idx = torch.where(ok, idx, torch.tensor(-1))
    return idx.reshape(64, KH, 2, 8 * I).to(torch.int32)


def _idx_packb(base: int, n: int) -> torch.Tensor:
    """Packed head bias [64]: b[d*8+co] = bias[co] (co < n), else 0."""
    co = torch.arange(8).repeat(8)
    idx = base + co
    idx[co >= n] = -1
    return idx.to(torch.int32)


class ShadowArena:
    """Builds and refreshes the arena for one FlatParamGroup + its module."""

    def __init__(self, group, module: torch.nn.Module):
        self.group = group
        dev = group.flat_param.device
        off = {p: o for p, (o, _) in zip(group.params, group._offsets)}

        pieces: List[torch.Tensor] = []           # int32 index chunks (cpu)
        plan: List[Tuple[torch.nn.Parameter, str, Tuple[int, ...], int]] = []
        total = 0
        for m in module.modules():
            if _conv_like(m):
                forms = (("p", _idx_p), ("tp", _idx_tp))
            elif _convt_like(m):
                forms = (("plain", _idx_plain), ("t", _idx_t))
            else:
                continue
            w = m.weight
            forms = list(forms)
            packed = (_conv_like(m) and head_packable(
                tuple(w.shape), m.stride, m.pad_mode, m.padding
                if not isinstance(m.padding, str) else (0,) * 4))
            if packed:
                forms.append(("packp", _idx_packp))
            for name, fn in forms:
                idx = fn(off[w], tuple(w.shape))
                pieces.append(idx.reshape(-1))
                plan.append((w, name, tuple(idx.shape), total))
                total += idx.numel()
            if m.bias is not None:
                idx = _idx_bias(off[m.bias], m.bias.numel())
                pieces.append(idx.reshape(-1))
                plan.append((m.bias, "bias_p", tuple(idx.shape), total))
                total += idx.numel()
                if packed:
                    idx = _idx_packb(off[m.bias], m.bias.numel())
                    pieces.append(idx.reshape(-1))
                    plan.append((m.bias, "packb", tuple(idx.shape), total))
                    total += idx.numel()

        pad = (-total) % 8
        if pad:
            pieces.append(torch.full((pad,), -1, dtype=torch.int32))
        self.idx = torch.cat(pieces).to(dev) if pieces else \
            torch.empty(0, dtype=torch.int32, device=dev)
        self.buf = torch.empty(self.idx.numel(), dtype=torch.bfloat16, device=dev)

        # per-(master, form) shaped views into the arena
        self.views: Dict[Tuple[int, str], torch.Tensor] = {}
        self._by_param: Dict[torch.Tensor, Dict[str, torch.Tensor]] = {}
        for p, name, shape, start in plan:
            v = self.buf[start:start + int(torch.tensor(shape).prod())].view(shape)
            self._by_param.setdefault(p, {})[name] = v

        self.refresh()

    def refresh(self):
        """One gather kernel: flat fp32 masters -> every bf16 form."""
        if self.idx.numel():
            backend.ext().shadow_gather(self.group.flat_param, self.idx, self.buf)

    def forms_of(self, p: torch.Tensor):
        return self._by_param.get(p)
