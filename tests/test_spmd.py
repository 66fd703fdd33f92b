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
