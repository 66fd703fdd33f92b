"""Property-based fuzz of the resharding math (hypothesis).

The slice engine is the correctness heart; these properties
pin it against brute force:

* overlap() agrees with torch boolean-mask intersection;
* any random disjoint tiling of a box assembles bit-exactly;
* extract_region ∘ region_view round-trips arbitrary sub-regions.
"""

import hypothesis.strategies as st
import torch
from hypothesis import given, settings

from torchstore_amd.ops.slicing import (
    assemble,
    extract_region,
    overlap,
    region_view,
    union_volume,
)

dims = st.integers(min_value=1, max_value=3)


@st.composite
def region_pair(draw):
    nd = draw(dims)
    shape = [draw(st.integers(1, 8)) for _ in range(nd)]
    def reg():
        off = [draw(st.integers(0, 6)) for _ in range(nd)]
        sz = [draw(st.integers(1, 8)) for _ in range(nd)]
        return tuple(off), tuple(sz)
    return shape, reg(), reg()


@given(region_pair())
@settings(max_examples=200, deadline=None)
def test_overlap_matches_bruteforce(data):
    _shape, (ao, asz), (bo, bsz) = data
    nd = len(ao)
    hi = [max(ao[d] + asz[d], bo[d] + bsz[d]) for d in range(nd)]
    a = torch.zeros(hi, dtype=torch.bool)
    b = torch.zeros(hi, dtype=torch.bool)
    a[tuple(slice(ao[d], ao[d] + asz[d]) for d in range(nd))] = True
    b[tuple(slice(bo[d], bo[d] + bsz[d]) for d in range(nd))] = True
    inter = a & b
    got = overlap(ao, asz, bo, bsz)
    if got is None:
        assert not inter.any()
    else:
        off, sz = got
        expect = torch.zeros_like(a)
        expect[tuple(slice(off[d], off[d] + sz[d]) for d in range(nd))] = True
        assert torch.equal(inter, expect)
        # union volume consistent with inclusion-exclusion
        vol = lambda s: int(torch.tensor(s).prod())
        assert union_volume([(ao, asz), (bo, bsz)]) == (
            vol(asz) + vol(bsz) - vol(sz)
        )


@st.composite
def random_tiling(draw):
    nd = draw(dims)
    shape = [draw(st.integers(2, 10)) for _ in range(nd)]
    # split each dim into 1-3 contiguous intervals -> grid tiling
    cuts = []
    for d in range(nd):
        n = draw(st.integers(1, min(3, shape[d])))
        pts = sorted(draw(
            st.lists(st.integers(1, shape[d] - 1), min_size=n - 1,
                     max_size=n - 1, unique=True)
        )) if n > 1 else []
        cuts.append([0] + pts + [shape[d]])
    return shape, cuts


@given(random_tiling())
@settings(max_examples=100, deadline=None)
def test_random_grid_tiling_assembles_exactly(data):
    import itertools

    shape, cuts = data
    full = torch.randn(shape)
    parts = []
    for cell in itertools.product(*(range(len(c) - 1) for c in cuts)):
        off = tuple(cuts[d][cell[d]] for d in range(len(shape)))
        sz = tuple(cuts[d][cell[d] + 1] - cuts[d][cell[d]]
                   for d in range(len(shape)))
        parts.append((off, extract_region(full, (0,) * len(shape), off, sz).clone()))
    out, origin = assemble(parts)
    assert origin == (0,) * len(shape)
    assert torch.equal(out, full)


@given(region_pair())
@settings(max_examples=150, deadline=None)
def test_region_view_roundtrip(data):
    _s, (doff, dsz), _b = data
    nd = len(doff)
    dest = torch.randn(dsz)
    # a random interior region of the dest (in GLOBAL coordinates)
    ro = tuple(min(doff[d] + dsz[d] - 1, doff[d] + d) for d in range(nd))
    rs = tuple(max(1, dsz[d] - (ro[d] - doff[d])) for d in range(nd))
    view = region_view(dest, doff, ro, rs)
    stamp = torch.randn(rs)
    view.copy_(stamp)
    again = extract_region(dest, doff, ro, rs)
    assert torch.equal(again, stamp)
