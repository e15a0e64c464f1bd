"""Property-based tests (hypothesis) for pure helper logic."""

import torch
from hypothesis import given, settings
from hypothesis import strategies as st


@given(st.integers(min_value=2, max_value=4096))
@settings(max_examples=200, deadline=None)
def test_default_ratios_product(total):
    """_default_ratios factorises any total: product == total, every
    ratio >= 2 (a ratio of 1 would be a no-op upsample block)."""
    from melgan_amd.config import _default_ratios

    ratios = _default_ratios(total)
    prod = 1
    for r in ratios:
        prod *= r
    assert prod == total
    assert all(r >= 2 for r in ratios)


@given(st.lists(st.sampled_from([(8, 4), (16, 2), (3, 3), (32, 1)]),
                min_size=1, max_size=6))
@settings(max_examples=50, deadline=None)
def test_load_flexible_any_rename(shapes):
    """Ordered shape-matching translation loads a state dict under ANY
    module-path renaming, for arbitrary layer stacks."""
    import torch.nn as nn

    from melgan_amd.engine.checkpoint import load_flexible

    def build():
        torch.manual_seed(0)
        return nn.Sequential(*[nn.Conv1d(c, c, k) for (c, k) in shapes])

    src = build()
    # rename every key prefix the way a foreign tree would
    state = {f"foreign.block{i}.conv.{kind}": v
             for k, v in src.state_dict().items()
             for i, kind in [(int(k.split(".")[0]), k.split(".")[-1])]}
    dst = build()
    with torch.no_grad():
        for p in dst.parameters():
            p.add_(1.0)  # make dst differ
    load_flexible(dst, state)
    for a, b in zip(dst.state_dict().values(), src.state_dict().values()):
        assert torch.equal(a, b)


@given(st.integers(min_value=1, max_value=200),
       st.integers(min_value=1, max_value=64))
@settings(max_examples=100, deadline=None)
def test_avg_pool_421_reference_shape(t_in, b):
    """The discriminator's pool output length matches torch for any T."""
    import torch.nn.functional as F

    x = torch.randn(1, 1, t_in + 3)
    y = F.avg_pool1d(x, kernel_size=4, stride=2, padding=1,
                     count_include_pad=False)
    assert y.size(2) == (x.size(2) + 2 - 4) // 2 + 1
