"""Shadow-arena index-map correctness (CPU: gather emulated in torch)."""
import torch

from cyclegan_amd.models import Generator, Discriminator
from cyclegan_amd.parallel import FlatParamGroup
from cyclegan_amd.ops.arena import ShadowArena
from cyclegan_amd.ops.shadow import _pad_dims


def _emulate_gather(arena, flat):
    """CPU emulation of shadow.hip's shadow_gather."""
    idx = arena.idx.long()
    return torch.where(idx >= 0, flat[idx.clamp(min=0)],
                       torch.zeros(())).to(torch.bfloat16)


def _slice(buf, view, arena):
    off = (view.data_ptr() - arena.buf.data_ptr()) // 2
    return buf[off:off + view.numel()].view(view.shape)


def _build(module, monkeypatch):
    group = FlatParamGroup(module)
    monkeypatch.setattr(ShadowArena, "refresh", lambda self: None)
    arena = ShadowArena(group, module)
    buf = _emulate_gather(arena, group.flat_param)
    return arena, buf


def test_arena_forms_match_legacy(monkeypatch):
    torch.manual_seed(0)
    for module in (Generator(num_residual_blocks=2), Discriminator()):
        arena, buf = _build(module, monkeypatch)
        for m in module.modules():
            name = type(m).__name__
            if name == "ConvNHWC":
                forms = arena.forms_of(m.weight)
                want_p = _pad_dims(m.weight.detach()).to(torch.bfloat16)
                assert torch.equal(_slice(buf, forms["p"], arena), want_p)
                want_tp = want_p.permute(3, 1, 2, 0).contiguous()
                assert torch.equal(_slice(buf, forms["tp"], arena), want_tp)
                if m.bias is not None:
                    want_b = torch.nn.functional.pad(
                        m.bias.detach(), (0, max(0, 8 - m.bias.numel()))
                    ).to(torch.bfloat16)
                    got_b = _slice(buf, arena.forms_of(m.bias)["bias_p"], arena)
                    assert torch.equal(got_b, want_b)
            elif name == "ConvTransposeNHWC":
                forms = arena.forms_of(m.weight)
                want = m.weight.detach().to(torch.bfloat16)
                assert torch.equal(_slice(buf, forms["plain"], arena), want)
                want_t = want.permute(3, 1, 2, 0).contiguous()
                assert torch.equal(_slice(buf, forms["t"], arena), want_t)


def test_arena_idx_shape_contract(monkeypatch):
    torch.manual_seed(0)
    m = Discriminator()
    arena, _ = _build(m, monkeypatch)
    assert arena.idx.numel() % 8 == 0
    assert arena.idx.numel() == arena.buf.numel()
    # every conv weight is covered
    n_conv = sum(1 for mm in m.modules()
                 if type(mm).__name__ in ("ConvNHWC", "ConvTransposeNHWC"))
    assert len(arena._by_param) >= n_conv
