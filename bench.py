"""Flagship benchmark: Llama-3-8B bf16 state_dict sync through the store.

Measures the BASELINE.json headline on MI355X: each rank holds the
FSDP-style Shard(0) layout of a random-init Llama-3-8B (~16.06 GB bf16),
one **step** = push the full state_dict into GPU-resident storage volumes
(put_state_dict) + pull it back in the TP-style layout (get_state_dict with
resharding).  value = whole-job aggregate GB/s moved (put+get bytes / step
time, max over ranks).  ``--mode direct`` instead measures the one-sided
weight-sync pull (a generator process reads the trainer's live parameters
over xGMI).  The reference (meta-pytorch/torchstore) publishes no numbers
 — this is the self-measured baseline.

Shards are expressed as :class:`LocalShard` (explicit TensorSlice layouts),
so the bench needs no process group: ranks coordinate through a TCPStore
(barriers + max-reduce).  torchrun only supplies the env.

Single GPU:      python bench.py --steps 5 --warmup 2
N GPUs (driver): python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
                     --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
"""

from __future__ import annotations

import argparse
import asyncio
import json
import os
import time

import torch

from torchstore_amd.runtime import Actor, endpoint


class GeneratorActor(Actor):
    """Serving-fleet stand-in for --mode direct: pulls weights one-sided
    over HIP IPC from the trainer's live parameter memory."""

    def __init__(self, controller, key, layers, world, rank, device_index):
        import torchstore_amd as ts
        from torchstore_amd.models import llama
        from torchstore_amd.strategy import LocalRankStrategy
        from torchstore_amd.weight_sync import DirectWeightSyncDest

        os.environ["RANK"] = str(rank)
        torch.cuda.set_device(device_index)
        ts.attach(controller, LocalRankStrategy())
        self.dst_sd = llama.make_local_shard_state_dict(
            rank, world, llama.tp_placement,
            device=f"cuda:{device_index}", layers=layers,
        )
        self.dest = DirectWeightSyncDest(ts.client(), key)

    @endpoint
    async def pull(self):
        t0 = time.perf_counter()
        await self.dest.pull(self.dst_sd)
        torch.cuda.synchronize()
        nbytes = sum(op.nbytes for op in self.dest._plan)
        return time.perf_counter() - t0, nbytes

    @endpoint
    def verify_pattern(self):
        """Every pulled entry must equal the position-determined pattern —
        verifies the cross-GPU read paths without exchanging data."""
        from torchstore_amd.models import llama
        from torchstore_amd.types import LocalShard

        bad = []
        for name, v in self.dst_sd.items():
            local = v.tensor if isinstance(v, LocalShard) else v
            offsets = (
                v.slice.offsets
                if isinstance(v, LocalShard)
                else (0,) * local.dim()
            )
            exp = llama.expected_pattern(
                tuple(local.shape), offsets, local.dtype, local.device
            )
            if not torch.equal(local, exp):
                bad.append(name)
        torch.cuda.synchronize()
        return bad


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=5)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--layers", type=int, default=None,
                   help="override layer count (debug only; invalidates the metric)")
    p.add_argument("--mode", choices=["state_dict", "direct"],
                   default="state_dict")
    return p.parse_args()


class Coord:
    """Rank coordination over a TCPStore: barriers + gather (no PG needed)."""

    def __init__(self, rank: int, world: int):
        self.rank = rank
        self.world = world
        self._n = 0
        self.store = None
        if world > 1:
            from datetime import timedelta

            from torch.distributed import TCPStore

            self.store = TCPStore(
                os.environ.get("MASTER_ADDR", "127.0.0.1"),
                int(os.environ.get("MASTER_PORT", "29500")) + 2,
                world, is_master=rank == 0,
                timeout=timedelta(seconds=300),
                wait_for_workers=True,
            )

    def barrier(self):
        if self.store is None:
            return
        self._n += 1
        key = f"bar/{self._n}"
        self.store.add(key, 1)
        while int(self.store.add(key, 0)) < self.world:
            time.sleep(0.001)

    def gather_floats(self, value: float):
        """Every rank contributes; rank 0 gets the list, others None."""
        if self.store is None:
            return [value]
        self._n += 1
        self.store.set(f"g/{self._n}/{self.rank}", repr(float(value)))
        self.barrier()
        if self.rank != 0:
            return None
        return [
            float(self.store.get(f"g/{self._n - 1}/{r}").decode())
            for r in range(self.world)
        ]


async def run_bench(args, rank, world, local_rank, coord: Coord):
    import torchstore_amd as ts
    from torchstore_amd.models import llama
    from torchstore_amd.strategy import LocalRankStrategy
    from torchstore_amd.types import LocalShard

    device = f"cuda:{local_rank}"
    layers = args.layers or llama.LAYERS
    src_sd = llama.make_local_shard_state_dict(
        rank, world, llama.fsdp_placement, device=device, layers=layers,
        pattern=True,
    )
    shapes = llama.llama3_8b_shapes(layers)
    payload_bytes = llama.total_bytes(shapes, torch.bfloat16)

    # store bring-up: rank 0 spawns volumes (one per GPU) + controller; the
    # SPMD bootstrap shares the controller handle through its rendezvous
    if world > 1:
        controller = await ts.initialize_spmd(
            strategy=LocalRankStrategy(), storage_device="auto"
        )
    else:
        controller = await ts.initialize(
            num_storage_volumes=1,
            strategy=LocalRankStrategy(),
            storage_device="auto",
        )

    def barrier():
        coord.barrier()
        torch.cuda.synchronize()

    gen_mesh = None
    pull_bytes = [0]
    phases = {"put": 0.0, "barrier": 0.0, "get": 0.0}
    if args.mode == "direct":
        # trainer = this process; generator = a separate process on the same
        # GPU (IPC handles cannot be opened by their exporting process)
        from torchstore_amd.runtime import spawn_actors

        await ts.put_state_dict(
            src_sd, "bench", direct=True, rank=rank, world_size=world
        )
        coord.barrier()
        gen_mesh = await asyncio.to_thread(
            spawn_actors, 1, GeneratorActor, f"generator-{rank}",
            controller, "bench", layers, world, rank, local_rank,
        )
        gen = gen_mesh.handles[0]

        async def one_step():
            # push = staging refresh (no cast here → sync only);
            # pull = one batched one-sided read into generator memory
            t0 = time.perf_counter()
            await ts.put_state_dict(src_sd, "bench", direct=True)
            t1 = time.perf_counter()
            _dt, pull_bytes[0] = await gen.pull.call_one()
            t2 = time.perf_counter()
            phases["put"] += t1 - t0
            phases["get"] += t2 - t1

    else:
        dst_sd = llama.make_local_shard_state_dict(
            rank, world, llama.tp_placement, device=device, layers=layers,
        )

        async def one_step():
            t0 = time.perf_counter()
            await ts.put_state_dict(src_sd, "bench")
            t1 = time.perf_counter()
            coord.barrier()  # all shards committed before the reshard pull
            t2 = time.perf_counter()
            await ts.get_state_dict("bench", dst_sd)
            t3 = time.perf_counter()
            phases["put"] += t1 - t0
            phases["barrier"] += t2 - t1
            phases["get"] += t3 - t2

    for _ in range(args.warmup):
        await one_step()
    barrier()

    # correctness guard (outside the timed region): every pulled shard must
    # equal the position-determined pattern — this validates the cross-GPU
    # (xGMI) read paths end to end, not just co-located regions
    if args.mode == "direct":
        bad = await gen.verify_pattern.call_one()
        if bad:
            raise RuntimeError(
                f"bench correctness probe failed (direct pull): {bad[:5]}"
            )
    else:
        bad = []
        for name, v in dst_sd.items():
            local = v.tensor if isinstance(v, LocalShard) else v
            offsets = (
                v.slice.offsets if isinstance(v, LocalShard)
                else (0,) * local.dim()
            )
            exp = llama.expected_pattern(
                tuple(local.shape), offsets, local.dtype, local.device
            )
            if not torch.equal(local, exp):
                bad.append(name)
        if bad:
            raise RuntimeError(
                f"bench correctness probe failed (reshard pull): {bad[:5]} "
                f"({len(bad)} of {len(dst_sd)} entries wrong)"
            )

    for k in phases:
        phases[k] = 0.0  # drop warmup contributions
    t0 = time.perf_counter()
    for _ in range(args.steps):
        await one_step()
    barrier()
    elapsed = time.perf_counter() - t0

    # per-rank phase breakdown (stderr): makes a bad SCALE curve diagnosable
    import sys as _sys

    print(
        f"[bench rank {rank}/{world}] per-step phases: "
        + " ".join(
            f"{k}={v / args.steps * 1e3:.1f}ms" for k, v in phases.items()
        ),
        file=_sys.stderr, flush=True,
    )

    times = coord.gather_floats(elapsed)
    if args.mode == "direct":
        rank_bytes = coord.gather_floats(float(pull_bytes[0]))
    else:
        rank_bytes = None

    if rank == 0:
        elapsed = max(times)
        ms_per_step = elapsed / args.steps * 1e3
        if args.mode == "direct":
            moved = int(sum(rank_bytes))  # actual bytes read per pull, all ranks
        else:
            moved = 2 * payload_bytes  # put + get per step, whole job
        gbps = moved / (elapsed / args.steps) / 1e9
        print(json.dumps({
            "metric": "llama3_8b_state_dict_sync_GBps",
            "value": round(gbps, 2),
            "unit": "GB/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic (Llama-3-8B shapes, position-pattern values so every rank verifies its resharded pull bit-exactly)",
            "config": {
                "model": "Llama-3-8B",
                "global_batch": None,
                "seq_len": None,
                "parallelism": f"store fsdp{world}->tp{world}",
                "payload_gb": round(payload_bytes / 1e9, 2),
                "mode": args.mode,
            },
        }), flush=True)

    if gen_mesh is not None:
        await gen_mesh.stop()
    coord.barrier()
    await ts.shutdown()  # collective in SPMD mode


def main():
    args = parse_args()
    if not torch.cuda.is_available():
        raise SystemExit("bench.py needs MI355X GPUs (torch.cuda unavailable)")
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    torch.cuda.set_device(local_rank % torch.cuda.device_count())
    coord = Coord(rank, world)
    try:
        asyncio.run(
            run_bench(args, rank, world,
                      local_rank % torch.cuda.device_count(), coord)
        )
    except Exception:
        import sys
        import traceback

        print(
            f"[bench rank {rank}/{world}] FAILED:\n{traceback.format_exc()}",
            file=sys.stderr, flush=True,
        )
        raise


if __name__ == "__main__":
    main()
