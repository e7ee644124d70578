"""fp8 delayed-scaling state: slot allocation semantics (CPU)."""
import torch

from cyclegan_amd.ops import fp8_state


def test_slots_are_stable_and_fresh_once():
    fp8_state.reset()
    w1 = torch.randn(4)
    w2 = torch.randn(4)
    p1, c1, fresh1 = fp8_state.slots_for(w1)
    assert fresh1 and p1.numel() == 1 and c1.numel() == 1
    p1b, c1b, fresh1b = fp8_state.slots_for(w1)
    assert not fresh1b
    assert p1b.data_ptr() == p1.data_ptr() and c1b.data_ptr() == c1.data_ptr()
    p2, _, fresh2 = fp8_state.slots_for(w2)
    assert fresh2 and p2.data_ptr() != p1.data_ptr()
    # slots live in one arena: prev/cur views a fixed stride apart
    assert abs(c1.data_ptr() - p1.data_ptr()) == fp8_state.CAP * 4
    fp8_state.reset()


def test_arena_values_zero_initialised():
    fp8_state.reset()
    w = torch.randn(3)
    p, c, _ = fp8_state.slots_for(w)
    assert float(p) == 0.0 and float(c) == 0.0
    p.fill_(2.5)
    p2, _, _ = fp8_state.slots_for(w)
    assert float(p2) == 2.5  # same storage
    fp8_state.reset()
