"""Property-based invariants (hypothesis) for the tile/halo geometry —
the arithmetic the reference embeds in 1500 lines of index tables
(spatial.py:177-335) and that every seam and exchange relies on."""

import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from mpi4dl_amd.ops.halo import (
    DIRS,
    TileLayout,
    _opposite,
    recv_region,
    send_region,
)

methods = st.sampled_from(["vertical", "horizontal", "square"])


def _layout(method, n):
    if method == "square":
        n = max(1, int(n**0.5)) ** 2
    return TileLayout(n, method), n


@given(methods, st.integers(1, 16), st.integers(1, 8), st.integers(1, 8))
@settings(max_examples=200, deadline=None)
def test_slice_input_partitions_image(method, n, th, tw):
    """Tiles partition the image exactly: every pixel in exactly one tile."""
    layout, n = _layout(method, n)
    H, W = layout.rows * th, layout.cols * tw
    x = torch.arange(H * W, dtype=torch.float32).reshape(1, 1, H, W)
    seen = torch.zeros(H, W)
    for t in range(n):
        tile = layout.slice_input(x, t)
        assert tile.shape[-2:] == (th, tw)
        r, c = layout.pos(t)
        seen[r * th : (r + 1) * th, c * tw : (c + 1) * tw] += 1
    assert (seen == 1).all()


@given(methods, st.integers(1, 16))
@settings(max_examples=100, deadline=None)
def test_neighbours_symmetric(method, n):
    """t' is t's neighbour in direction d  <=>  t is t''s in -d."""
    layout, n = _layout(method, n)
    for t in range(n):
        for d, nb in layout.neighbours(t):
            back = dict(layout.neighbours(nb))
            assert back.get(_opposite(d)) == t, (t, d, nb)


@given(
    st.sampled_from(DIRS),
    st.integers(1, 32),
    st.integers(1, 32),
    st.integers(0, 4),
    st.integers(0, 4),
)
@settings(max_examples=200, deadline=None)
def test_send_recv_regions_match(d, H, W, hh, hw):
    """What I send toward d is exactly the shape the d-neighbour receives
    from -d, and both lie inside the padded tensor. Only valid when the
    halo fits the tile (hh <= H, hw <= W) — exchange_padded asserts
    exactly that at runtime."""
    from hypothesis import assume

    assume(hh <= H and hw <= W)
    h = (hh, hw)
    srs, scs = send_region(d, H, W, h)
    rrs, rcs = recv_region(_opposite(d), H, W, h)
    assert srs[1] - srs[0] == rrs[1] - rrs[0]
    assert scs[1] - scs[0] == rcs[1] - rcs[0]
    Hp, Wp = H + 2 * hh, W + 2 * hw
    for (a, b), lim in [(srs, Hp), (scs, Wp), (rrs, Hp), (rcs, Wp)]:
        assert 0 <= a <= b <= lim
    # send regions live in the interior (nominal region)
    assert srs[0] >= hh and srs[1] <= hh + H
    assert scs[0] >= hw and scs[1] <= hw + W


@given(
    st.sampled_from(DIRS),
    st.integers(1, 16),
    st.integers(1, 16),
    st.integers(1, 3),
    st.integers(1, 3),
)
@settings(max_examples=200, deadline=None)
def test_recv_regions_tile_the_ring(d, H, W, hh, hw):
    """Each direction's recv region is disjoint from every other
    direction's (no double-writes in the pad ring)."""
    h = (hh, hw)
    Hp, Wp = H + 2 * hh, W + 2 * hw
    mask = torch.zeros(Hp, Wp)
    for dd in DIRS:
        (r0, r1), (c0, c1) = recv_region(dd, H, W, h)
        mask[r0:r1, c0:c1] += 1
    assert (mask <= 1).all(), mask
