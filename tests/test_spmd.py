"""SPMD (torchrun-style) lifecycle: collective bring-up, cross-rank
put/get, collective shutdown — 2 OS processes on loopback."""

import asyncio
import json
import multiprocessing as mp
import os
import tempfile

import pytest
import torch

from torchstore_amd.utils.net import pick_free_port


def _spmd_worker(rank, world, master_port, result_path, strategy_name="local_rank"):
    os.environ.update(
        {
            "RANK": str(rank),
            "LOCAL_RANK": str(rank),
            "WORLD_SIZE": str(world),
            "LOCAL_WORLD_SIZE": str(world),
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(master_port),
        }
    )
    import torchstore_amd as ts

    strategy = {
        "local_rank": ts.LocalRankStrategy,
        "host": ts.HostStrategy,
    }[strategy_name]()

    async def main():
        await ts.initialize_spmd(strategy=strategy)
        t = torch.full((16,), float(rank))
        await ts.put(f"rank{rank}/data", t)
        # cross-rank visibility: wait for the peer's key
        peer = (rank + 1) % world
        for _ in range(100):
            if await ts.exists(f"rank{peer}/data"):
                break
            await asyncio.sleep(0.1)
        out = await ts.get(f"rank{peer}/data")
        ok = bool(out.eq(float(peer)).all())
        ks = sorted(await ts.keys())
        await ts.shutdown()
        return {"ok": ok, "keys": ks}

    try:
        result = asyncio.run(main())
    except Exception as exc:  # noqa: BLE001
        import traceback

        result = {"error": f"{exc}\n{traceback.format_exc()}"}
    with open(result_path, "w") as f:
        json.dump(result, f)


@pytest.mark.parametrize("strategy_name", ["local_rank", "host"])
def test_spmd_lifecycle(strategy_name):
    world = 2
    port = pick_free_port()
    ctx = mp.get_context("spawn")
    procs = []
    paths = []
    for rank in range(world):
        path = tempfile.mktemp(prefix=f"spmd-res-{rank}")
        paths.append(path)
        p = ctx.Process(
            target=_spmd_worker,
            args=(rank, world, port, path, strategy_name),
        )
        p.start()
        procs.append(p)
    for p in procs:
        p.join(timeout=180)
        assert not p.is_alive(), "spmd worker hung"
    for rank, path in enumerate(paths):
        with open(path) as f:
            result = json.load(f)
        assert "error" not in result, f"rank {rank}: {result.get('error')}"
        assert result["ok"], f"rank {rank} got wrong peer data"
        assert result["keys"] == ["rank0/data", "rank1/data"]


def test_spmd_env_parsing():
    from torchstore_amd.spmd import SPMDEnv

    env = SPMDEnv.from_env(
        {
            "RANK": "3",
            "WORLD_SIZE": "8",
            "LOCAL_RANK": "1",
            "LOCAL_WORLD_SIZE": "2",
            "MASTER_ADDR": "10.0.0.1",
            "MASTER_PORT": "1234",
        }
    )
    assert env.rank == 3 and env.num_hosts == 4
    with pytest.raises(RuntimeError, match="missing"):
        SPMDEnv.from_env({})


def _multihost_worker(rank, world, local_world, master_port, result_path,
                      strategy_name):
    """Each process fakes being a different HOST (HOSTNAME env) with
    local_world ranks per host — exercises the per-host volume spawn path
    (reference capability: Monarch host-mesh spawning, spmd.py:317-326)."""
    host_index = rank // local_world
    os.environ.update(
        {
            "RANK": str(rank),
            "LOCAL_RANK": str(rank % local_world),
            "WORLD_SIZE": str(world),
            "LOCAL_WORLD_SIZE": str(local_world),
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(master_port),
            "HOSTNAME": f"fakehost{host_index}",
        }
    )
    import torchstore_amd as ts

    strategy = {
        "local_rank": ts.LocalRankStrategy,
        "host": ts.HostStrategy,
    }[strategy_name]()

    async def main():
        controller = await ts.initialize_spmd(strategy=strategy)
        c = ts.client()
        volumes = await c._ensure_volumes()
        hostnames = sorted({v.hostname for v in volumes.values()})
        volume_ids = sorted(volumes.keys())
        # every rank writes; every rank reads the peer host's key
        await ts.put(f"r{rank}", torch.full((8,), float(rank)))
        peer = (rank + local_world) % world  # a rank on the OTHER host
        for _ in range(200):
            if await ts.exists(f"r{peer}"):
                break
            await asyncio.sleep(0.05)
        out = await ts.get(f"r{peer}")
        ok = bool(out.eq(float(peer)).all())
        await ts.shutdown()
        return {
            "ok": ok,
            "hostnames": hostnames,
            "volume_ids": volume_ids,
        }

    try:
        result = asyncio.run(main())
    except Exception as exc:  # noqa: BLE001
        import traceback

        result = {"error": f"{exc}\n{traceback.format_exc()}"}
    with open(result_path, "w") as f:
        json.dump(result, f)


@pytest.mark.parametrize("strategy_name", ["host", "local_rank"])
def test_spmd_multihost_volume_placement(strategy_name):
    """2 fake hosts x 1 rank: each host's local-rank-0 spawns its own
    volumes; the controller sees volumes on BOTH hostnames."""
    world, local_world = 2, 1
    port = pick_free_port()
    ctx = mp.get_context("spawn")
    procs, paths = [], []
    for rank in range(world):
        path = tempfile.mktemp(prefix=f"spmd-mh-{rank}")
        paths.append(path)
        p = ctx.Process(
            target=_multihost_worker,
            args=(rank, world, local_world, port, path, strategy_name),
        )
        p.start()
        procs.append(p)
    for p in procs:
        p.join(timeout=240)
        assert not p.is_alive(), "multihost spmd worker hung"
    for rank, path in enumerate(paths):
        with open(path) as f:
            result = json.load(f)
        assert "error" not in result, f"rank {rank}: {result.get('error')}"
        assert result["ok"], f"rank {rank} read wrong peer data"
        assert result["hostnames"] == ["fakehost0", "fakehost1"], result
        if strategy_name == "host":
            assert result["volume_ids"] == ["fakehost0", "fakehost1"]
        else:
            assert result["volume_ids"] == ["0", "1"]


@pytest.mark.skipif(
    os.environ.get("TORCHSTORE_AMD_SLOW_TESTS", "0") != "1",
    reason="4-process bring-up is slow; TORCHSTORE_AMD_SLOW_TESTS=1 enables",
)
def test_spmd_two_hosts_two_ranks_each():
    """2 fake hosts x 2 ranks: per-host volume spawns with global id
    offsets (rank_offset seed) + cross-host gets."""
    world, local_world = 4, 2
    port = pick_free_port()
    ctx = mp.get_context("spawn")
    procs, paths = [], []
    for rank in range(world):
        path = tempfile.mktemp(prefix=f"spmd-mh4-{rank}")
        paths.append(path)
        p = ctx.Process(
            target=_multihost_worker,
            args=(rank, world, local_world, port, path, "local_rank"),
        )
        p.start()
        procs.append(p)
    for p in procs:
        p.join(timeout=300)
        assert not p.is_alive(), "worker hung"
    for rank, path in enumerate(paths):
        with open(path) as f:
            result = json.load(f)
        assert "error" not in result, f"rank {rank}: {result.get('error')}"
        assert result["ok"], f"rank {rank} read wrong peer data"
        assert result["volume_ids"] == ["0", "1", "2", "3"], result
        assert result["hostnames"] == ["fakehost0", "fakehost1"], result
