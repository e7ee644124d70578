"""bf16 shadow-weight cache semantics (cyclegan_amd/ops/shadow.py).

The GPU compute path depends on these invariants; the round-1
stale-shadow bug (fused Adam bypassing version counters) is the failure
mode they guard against. All CPU-testable: the caches key on tensor
identity + version, not device.
"""

import torch

from cyclegan_amd.ops import shadow


def _like_bf16():
    return torch.empty(1, dtype=torch.bfloat16)


def test_shadow_caches_per_version():
    w = torch.randn(4, 3, 3, 4)
    s1 = shadow.compute_weight(w, _like_bf16())
    s2 = shadow.compute_weight(w, _like_bf16())
    assert s1 is s2, "same version must hit the cache"
    assert s1.dtype == torch.bfloat16


def test_shadow_refreshes_on_inplace_update():
    w = torch.randn(4, 3, 3, 4)
    s1 = shadow.compute_weight(w, _like_bf16())
    with torch.no_grad():
        w.add_(1.0)  # dispatcher op: bumps the version counter
    s2 = shadow.compute_weight(w, _like_bf16())
    assert s2 is not s1
    assert torch.equal(s2, w.detach().to(torch.bfloat16))


def test_shadow_stale_without_version_bump_then_explicit_bump():
    """A raw (dispatcher-bypassing) mutation leaves the cache stale —
    exactly what the fused Adam kernel does — until the version counter
    is bumped explicitly (FlatParamGroup.bump_versions)."""
    w = torch.randn(4, 3, 3, 4)
    s1 = shadow.compute_weight(w, _like_bf16())
    w.detach().numpy()[:] += 1.0  # mutate storage without a version bump
    s2 = shadow.compute_weight(w, _like_bf16())
    assert s2 is s1, "cache cannot see dispatcher-bypassing writes"
    torch.autograd.graph.increment_version(w)
    s3 = shadow.compute_weight(w, _like_bf16())
    assert s3 is not s1
    assert torch.equal(s3, w.detach().to(torch.bfloat16))


def test_bump_versions_reaches_params_repointed_at_flat_views():
    """`p.data = view` keeps the param's own version counter — bumping
    the flat buffer does NOT propagate (the root cause of the round-1
    bug); FlatParamGroup.bump_versions must bump each param."""
    from cyclegan_amd.parallel import FlatParamGroup
    m = torch.nn.Linear(3, 3)
    grp = FlatParamGroup(m)
    p = grp.params[0]
    v0 = p._version
    torch.autograd.graph.increment_version(grp.flat_param)
    assert p._version == v0, "flat-buffer bump must not reach the param"
    grp.bump_versions()
    assert p._version > v0


def test_padded_shadow_zero_channels_inert():
    w = torch.randn(3, 3, 3, 3)  # Cout=3, Cin=3 -> padded to 8
    sp = shadow.compute_weight_p(w, _like_bf16())
    assert sp.shape == (8, 3, 3, 8)
    assert torch.equal(sp[:3, :, :, :3].float(),
                       w.detach().to(torch.bfloat16).float())
    assert sp[3:].abs().sum() == 0 and sp[:, :, :, 3:].abs().sum() == 0
