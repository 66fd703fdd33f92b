"""Flagship benchmark: Llama-3-8B bf16 state_dict sync through the store.

Measures the BASELINE.json headline on MI355X: each rank holds the
FSDP-style Shard(0) layout of a random-init Llama-3-8B (~16.06 GB bf16),
one **step** = push the full state_dict into GPU-resident storage volumes
(put_state_dict) + pull it back in the TP-style layout (get_state_dict with
resharding).  value = whole-job aggregate GB/s moved (put+get bytes / step
time, max over ranks).  The reference (meta-pytorch/torchstore) publishes
no numbers — this is the self-measured baseline.

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
    def verify(self, name, expected):
        from torchstore_amd.types import LocalShard

        v = self.dst_sd[name]
        local = v.tensor if isinstance(v, LocalShard) else v
        return torch.equal(local, expected.to(local.device))


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


def setup_dist(args):
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if world > 1:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29531")
        torch.distributed.init_process_group(
            "nccl", rank=rank, world_size=world
        )
        torch.cuda.set_device(local_rank)
    else:
        if torch.cuda.is_available():
            torch.cuda.set_device(0)
    return rank, world, local_rank


async def run_bench(args, rank, world, local_rank):
    import torchstore_amd as ts
    from torchstore_amd.models import llama
    from torchstore_amd.strategy import LocalRankStrategy

    device = f"cuda:{local_rank}"
    mesh = None
    if world > 1:
        from torch.distributed.device_mesh import init_device_mesh

        mesh = init_device_mesh("cuda", (world,))

    layers = args.layers or llama.LAYERS
    src_sd = llama.make_sharded_state_dict(
        mesh, llama.fsdp_placement, device=device, layers=layers, seed=1234 + rank
    )
    dst_sd = llama.make_sharded_state_dict(
        mesh, llama.tp_placement, device=device, layers=layers, zero=True
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
        if world > 1:
            torch.distributed.barrier()
        torch.cuda.synchronize()

    if args.mode == "direct":
        # trainer = this process; generator = a separate process on the same
        # GPU (IPC handles cannot be opened by their exporting process)
        from torchstore_amd.runtime import spawn_actors

        await ts.put_state_dict(
            src_sd, "bench", direct=True, rank=rank, world_size=world
        )
        if world > 1:
            torch.distributed.barrier()
        gen_mesh = await asyncio.to_thread(
            spawn_actors, 1, GeneratorActor, f"generator-{rank}",
            controller, "bench", layers, world, rank, local_rank,
        )
        gen = gen_mesh.handles[0]
        pull_bytes = [0]

        async def one_step():
            # push = staging refresh (no cast here → sync only);
            # pull = one batched one-sided read into generator memory
            await ts.put_state_dict(src_sd, "bench", direct=True)
            _dt, pull_bytes[0] = await gen.pull.call_one()

    else:

        async def one_step():
            await ts.put_state_dict(src_sd, "bench")
            if world > 1:
                torch.distributed.barrier()  # all shards committed
            await ts.get_state_dict("bench", dst_sd)

    for _ in range(args.warmup):
        await one_step()
    barrier()
    # correctness guard (outside the timed region): the norm weight has the
    # same layout in both placements, so the pulled local shard must equal
    # the pushed one bit-for-bit
    probe = "model.norm.weight"
    src_t = src_sd[probe]
    put_local = getattr(src_t, "to_local", lambda: src_t)
    if args.mode == "direct":
        ok = await gen.verify.call_one(probe, put_local().cpu())
        if not ok:
            raise RuntimeError("bench correctness probe failed (direct pull)")
    else:
        dst_t = dst_sd[probe]
        get_local = getattr(dst_t, "to_local", lambda: dst_t)
        if not torch.equal(get_local(), put_local()):
            raise RuntimeError("bench correctness probe failed: pulled != pushed")

    t0 = time.perf_counter()
    for _ in range(args.steps):
        await one_step()
    barrier()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if world > 1:
        t = torch.tensor([elapsed], device=device)
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = t.item()

    ms_per_step = elapsed / args.steps * 1e3
    if args.mode == "direct":
        moved = pull_bytes[0]  # actual bytes read per pull, this rank
        if world > 1:
            t = torch.tensor([float(moved)], device=device)
            torch.distributed.all_reduce(t)
            moved = int(t.item())
    else:
        moved = 2 * payload_bytes  # put + get per step, whole job
    gbps = moved / (elapsed / args.steps) / 1e9

    if rank == 0:
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
            "data": "synthetic (random-init Llama-3-8B shapes)",
            "config": {
                "model": "Llama-3-8B",
                "global_batch": None,
                "seq_len": None,
                "parallelism": f"store fsdp{world}->tp{world}",
                "payload_gb": round(payload_bytes / 1e9, 2),
                "mode": args.mode,
            },
        }))

    if args.mode == "direct":
        await gen_mesh.stop()
    if world > 1:
        torch.distributed.barrier()
    await ts.shutdown()  # collective in SPMD mode
    if world > 1:
        torch.distributed.destroy_process_group()


def main():
    args = parse_args()
    if not torch.cuda.is_available():
        raise SystemExit("bench.py needs MI355X GPUs (torch.cuda unavailable)")
    rank, world, local_rank = setup_dist(args)
    try:
        asyncio.run(run_bench(args, rank, world, local_rank))
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
