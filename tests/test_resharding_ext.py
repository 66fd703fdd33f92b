"""Extended resharding matrix, behind a slow gate.

Mirrors the reference's extended matrix
(/root/reference/tests/test_resharding_ext.py:29-133): the full
shrink/grow × sharding-dim product, 2-D↔2-D dim permutations, and 3-D
tensors.  Enable with ``TORCHSTORE_AMD_SLOW_TESTS=1`` (the reference gates
on TORCHSTORE_ENABLE_SLOW_TESTS the same way).
"""

import os

import pytest

from tests.test_resharding import _reshard_case

requires_slow = pytest.mark.skipif(
    os.environ.get("TORCHSTORE_AMD_SLOW_TESTS", "0") != "1",
    reason="slow matrix disabled; TORCHSTORE_AMD_SLOW_TESTS=1 enables",
)


@requires_slow
@pytest.mark.parametrize(
    "put_world,get_world,put_dim,get_dim",
    [
        # shrink, every dim pairing
        (4, 2, 0, 0),
        (4, 2, 0, 1),
        (4, 2, 1, 0),
        (4, 2, 1, 1),
        # grow, every dim pairing
        (2, 4, 0, 0),
        (2, 4, 0, 1),
        (2, 4, 1, 0),
        (2, 4, 1, 1),
    ],
)
async def test_1d_matrix(put_world, get_world, put_dim, get_dim):
    await _reshard_case(
        put_world, (put_world,), [str(put_dim)],
        get_world, (get_world,), [str(get_dim)],
    )


@requires_slow
@pytest.mark.parametrize(
    "put_dims,get_dims",
    [
        ((1, 1), (0, 1)),
        ((1, 0), (1, 0)),
        ((0, 0), (0, 1)),
        ((1, 1), (0, 0)),
    ],
)
async def test_2d_to_2d_matrix(put_dims, get_dims):
    await _reshard_case(
        4, (2, 2), [str(put_dims[0]), str(put_dims[1])],
        4, (2, 2), [str(get_dims[0]), str(get_dims[1])],
    )


@requires_slow
@pytest.mark.parametrize(
    "put_dim,get_dim",
    [(0, 2), (2, 0), (1, 2)],
)
async def test_3d_tensor_reshard(put_dim, get_dim):
    """3-D payloads: the slice engine is dimension-agnostic."""
    await _reshard_case(
        2, (2,), [str(put_dim)],
        2, (2,), [str(get_dim)],
        shape=(8, 12, 10),
    )


@requires_slow
@pytest.mark.parametrize("put_dim", [0, 1])
async def test_replicate_matrix(put_dim):
    """Shard→Replicate and Replicate→Shard across world sizes (the
    reference's TODO — 'test Replicate as well')."""
    await _reshard_case(4, (4,), [str(put_dim)], 2, (2,), ["r"])
    await _reshard_case(2, (2,), ["r"], 4, (4,), [str(put_dim)])
