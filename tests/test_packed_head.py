"""Packed head-conv math (ops.conv._ConvHeadPackedFn) validated on CPU
with pure torch: the 8-pixel packing + channel folding + weight fold-back
must reproduce the direct 7x7 reflect conv exactly."""
import torch
import torch.nn.functional as F

from cyclegan_amd.ops.conv import _conv_ref, _fold_packed_dw
from cyclegan_amd.ops.shadow import _pack_head_weight_torch
from cyclegan_amd.ops import arena as arena_mod


def _packed_forward(x, w, bias=None):
    """Torch emulation of the packed path (fp32)."""
    B, H, W, Cin = x.shape
    xp = F.pad(x.permute(0, 3, 1, 2), (3, 5, 3, 3), mode="reflect")
    xp = xp.permute(0, 2, 3, 1).contiguous()            # [B,H+6,W+8,Cin]
    xf = xp.view(B, H + 6, (W + 8) // 8, 8 * Cin)
    wpk = _pack_head_weight_torch(w)
    bpk = None
    if bias is not None:
        n = bias.numel()
        bp = F.pad(bias, (0, 8 - n)) if n < 8 else bias
        bpk = bp.repeat(8)
    yp = _conv_ref(xf, wpk, bpk, 1, (0, 0, 0, 0), "zeros")
    return yp.reshape(B, H, W, 8)[..., : w.shape[0]]


def test_packed_forward_matches_direct():
    torch.manual_seed(0)
    for W in (16, 24):
        x = torch.randn(2, 16, W, 8)
        w = torch.randn(3, 7, 7, 8) * 0.2
        b = torch.randn(3) * 0.1
        y_packed = _packed_forward(x, w, b)
        y_direct = _conv_ref(x, w, b, 1, (3, 3, 3, 3), "reflect")
        assert torch.allclose(y_packed, y_direct, atol=1e-4), \
            (y_packed - y_direct).abs().max()


def test_packed_wgrad_fold_matches_direct():
    torch.manual_seed(1)
    x = torch.randn(2, 16, 16, 8)
    w = torch.randn(3, 7, 7, 8) * 0.2

    wg1 = w.clone().requires_grad_(True)
    y1 = _packed_forward(x, wg1)
    dy = torch.randn_like(y1)
    y1.backward(dy)

    wg2 = w.clone().requires_grad_(True)
    y2 = _conv_ref(x, wg2, None, 1, (3, 3, 3, 3), "reflect")
    y2.backward(dy)

    assert torch.allclose(wg1.grad, wg2.grad, atol=1e-4)


def test_fold_packed_dw_inverts_packing():
    """_fold_packed_dw of a packed-conv wgrad: check by building a packed
    'gradient' whose entries are the packed copies of a known dw — the
    fold must sum the 7 aliased copies of each tap back to 7x the value
    only where copies exist; instead verify against autograd equality
    through the torch emulation."""
    torch.manual_seed(2)
    O, KH, KW, I = 3, 7, 7, 8
    dw_true = torch.randn(8, KH, 2, 8, I)  # pretend packed grads [d? ...]
    # direct identity check: pack a weight, fold its 'gradient pattern'
    w = torch.randn(O, KH, KW, I)
    wpk = _pack_head_weight_torch(w)       # [64,KH,2,8I]
    folded = _fold_packed_dw(wpk, (O, KH, KW, I))
    # every tap appears once per d in range -> fold sums 8 copies where
    # 0 <= tx+d <= 14 i.e. all 8 d values for every tx in 0..6
    assert torch.allclose(folded, w * 8.0, atol=1e-5)


def test_arena_packp_matches_torch_pack(monkeypatch):
    from cyclegan_amd.models import Generator
    from cyclegan_amd.parallel import FlatParamGroup
    from cyclegan_amd.ops.arena import ShadowArena
    torch.manual_seed(3)
    gmod = Generator(num_residual_blocks=1)
    group = FlatParamGroup(gmod)
    monkeypatch.setattr(ShadowArena, "refresh", lambda self: None)
    a = ShadowArena(group, gmod)
    idx = a.idx.long()
    buf = torch.where(idx >= 0, group.flat_param[idx.clamp(min=0)],
                      torch.zeros(())).to(torch.bfloat16)
    head = gmod.head
    forms = a.forms_of(head.weight)
    assert "packp" in forms, "head conv must register the packed form"
    off = (forms["packp"].data_ptr() - a.buf.data_ptr()) // 2
    got = buf[off:off + forms["packp"].numel()].view(forms["packp"].shape)
    want = _pack_head_weight_torch(head.weight.detach()).to(torch.bfloat16)
    assert torch.equal(got, want)
    bforms = a.forms_of(head.bias)
    assert "packb" in bforms
    offb = (bforms["packb"].data_ptr() - a.buf.data_ptr()) // 2
    gotb = buf[offb:offb + 64]
    n = head.bias.numel()
    wantb = torch.nn.functional.pad(head.bias.detach(), (0, 8 - n)) \
        .repeat(8).to(torch.bfloat16)
    assert torch.equal(gotb, wantb)
