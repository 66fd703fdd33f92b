"""Stability soak: repeated mixed traffic against a GPU-resident store.

Catches slow leaks the unit suites can't see: IPC handle caches, SHM
segments, staging pools, plan caches, allocator growth.  Prints client
HBM + volume stats every ``--report`` iterations and asserts the
steady-state memory watermark stops growing after warmup.

    python benchmarks/soak.py --iters 200
"""

from __future__ import annotations

import argparse
import asyncio
import os
import random
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import torchstore_amd as ts
from torchstore_amd.strategy import LocalRankStrategy
from torchstore_amd.types import LocalShard, TensorSlice


async def run(iters: int, report: int):
    await ts.initialize(
        num_storage_volumes=2,
        strategy=LocalRankStrategy(),
        storage_device="auto",
    )
    rng = random.Random(7)
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    watermark_after_warmup = None
    t0 = time.perf_counter()
    try:
        for i in range(iters):
            # mixed batch: a few tensors of varying sizes + an object
            items = {
                f"s/{i % 5}/a": torch.randn(
                    rng.choice([1 << 16, 1 << 20, 1 << 22]), device=dev
                ),
                f"s/{i % 5}/b": torch.randn(256, 513, device=dev),
                f"s/{i % 5}/meta": {"iter": i},
            }
            await ts.put_batch(items)
            # sharded put under a rotating epoch (exercises epoch release)
            world = 2 if i % 2 == 0 else 4
            for r in range(world):
                os.environ["RANK"] = str(r)
                rows = 64 // world
                await ts.put(
                    f"s/{i % 3}/w",
                    LocalShard(
                        tensor=torch.full((rows, 32), float(i), device=dev),
                        slice=TensorSlice(
                            offsets=(r * rows, 0), local_shape=(rows, 32),
                            global_shape=(64, 32), coordinates=(r,),
                            mesh_shape=(world,),
                        ),
                    ),
                )
            out = await ts.get(f"s/{i % 3}/w")
            assert out.eq(float(i)).all()
            got = await ts.get_batch(
                {k: torch.zeros_like(v) for k, v in items.items()
                 if isinstance(v, torch.Tensor)}
            )
            for k, v in got.items():
                assert torch.equal(v, items[k]), k
            if i % 7 == 6:
                await ts.delete_batch(list(items.keys()))
            if dev == "cuda" and i % 50 == 49:
                # periodic >=2 GiB traffic: auto-split + direct push/pull
                # + epoch replacement + delete reclamation
                huge = torch.empty(560_000_000, dtype=torch.float32,
                                   device=dev)
                huge.fill_(float(i))
                await ts.put("s/huge", huge)
                back = torch.zeros_like(huge)
                await ts.get("s/huge", back)
                assert back[::1_000_003].eq(float(i)).all()
                del huge, back
                if i % 100 == 99:
                    await ts.delete("s/huge")
            if (i + 1) % report == 0:
                torch.cuda.synchronize() if dev == "cuda" else None
                used = (
                    torch.cuda.memory_allocated() if dev == "cuda" else 0
                )
                stats = await ts.client()._controller.stats.call_one()
                print(
                    f"iter {i + 1}/{iters}: client_alloc={used / 1e6:.0f} MB "
                    f"controller={stats} "
                    f"({(i + 1) / (time.perf_counter() - t0):.1f} it/s)",
                    flush=True,
                )
                if i + 1 == 2 * report:
                    watermark_after_warmup = used
                elif watermark_after_warmup is not None:
                    # allow 256 MB of jitter (allocator caching, varying
                    # tensor sizes) — catches monotonic leaks
                    assert used < watermark_after_warmup + (256 << 20), (
                        f"client memory grew past warmup watermark: "
                        f"{used} vs {watermark_after_warmup}"
                    )
        print("soak ok", flush=True)
    finally:
        await ts.shutdown()


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=200)
    p.add_argument("--report", type=int, default=25)
    args = p.parse_args()
    asyncio.run(run(args.iters, args.report))


if __name__ == "__main__":
    main()
