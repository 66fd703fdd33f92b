"""SPMD usage under torchrun: every rank calls initialize_spmd; rank 0
spawns the store, the others attach through the rendezvous.

Run (CPU demo, any machine):
    torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 2 \
        example/spmd_torchrun.py

On a GPU node the volumes become HBM-resident automatically
(`storage_device="auto"`).
"""

import asyncio
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import torchstore_amd as ts


async def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    device = "cuda" if torch.cuda.is_available() else "cpu"

    await ts.initialize_spmd(
        strategy=ts.LocalRankStrategy(),
        storage_device="auto" if device == "cuda" else "cpu",
    )

    # every rank publishes its own tensor...
    await ts.put(f"stats/rank{rank}", torch.full((8,), float(rank), device=device))

    # ...and reads a peer's (cross-process, cross-volume)
    peer = (rank + 1) % world
    for _ in range(200):
        if await ts.exists(f"stats/rank{peer}"):
            break
        await asyncio.sleep(0.05)
    out = await ts.get(f"stats/rank{peer}")
    assert out.eq(float(peer)).all()
    print(f"rank {rank}: read peer {peer}'s tensor ok "
          f"(keys: {sorted(await ts.keys('stats'))})")

    await ts.shutdown()  # collective: barriers + rank-0 teardown


if __name__ == "__main__":
    asyncio.run(main())
