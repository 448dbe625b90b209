"""Property-based tests (hypothesis): structural invariants of the arena
and the episode sampler."""

import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from howtotrainyourmamlpytorch_amd.models.arena import ParamArena


@given(st.lists(st.tuples(st.integers(1, 5), st.integers(1, 5)),
                min_size=1, max_size=6))
@settings(max_examples=30, deadline=None)
def test_arena_pack_views_roundtrip(shapes):
    named_shapes = [(f"p{i}", s) for i, s in enumerate(shapes)]
    arena = ParamArena(named_shapes)
    named = {f"p{i}": torch.randn(*s) for i, s in enumerate(shapes)}
    flat = arena.pack(named)
    assert flat.numel() == sum(a * b for a, b in shapes)
    views = arena.views(flat)
    for name, t in named.items():
        torch.testing.assert_close(views[name], t)
    # 2-D views cover the arena disjointly
    t2 = flat.unsqueeze(0).repeat(3, 1)
    v2 = arena.views(t2)
    for i, s in enumerate(shapes):
        assert v2[f"p{i}"].shape == (3, *s)
    # slot index maps every element to its spec
    idx = arena.slot_index()
    for spec in arena.specs:
        assert (idx[spec.offset:spec.offset + spec.numel] == spec.slot).all()


@given(st.integers(0, 10_000), st.integers(2, 5), st.integers(1, 3),
       st.integers(1, 3))
@settings(max_examples=25, deadline=None)
def test_synthetic_episode_seed_purity(seed, ways, shots, targets):
    from howtotrainyourmamlpytorch_amd.config import get_args
    from howtotrainyourmamlpytorch_amd.data import SyntheticEpisodeStream
    args = get_args([
        "--num_classes_per_set", str(ways),
        "--num_samples_per_class", str(shots),
        "--num_target_samples", str(targets),
        "--image_height", "8", "--image_width", "8", "--image_channels", "1",
        "--batch_size", "2",
    ])
    s = SyntheticEpisodeStream(args)
    a = s._episode(seed)
    b = s._episode(seed)
    for x, y in zip(a, b):
        torch.testing.assert_close(x, y, rtol=0, atol=0)
    assert a[0].shape == (ways, shots, 1, 8, 8)
    assert a[2].max().item() == ways - 1
