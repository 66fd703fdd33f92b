"""Capacity benchmark: many-GB put_batch/get_batch across GPU-resident
volumes (the BASELINE '200 GB across 8 volumes' config, scaled to the
available GPUs — 288 GB HBM per MI355X holds client + volume copies).

    python benchmarks/capacity.py --total-gb 100 --chunk-mb 1024
"""

from __future__ import annotations

import argparse
import asyncio
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import torchstore_amd as ts
from torchstore_amd.strategy import LocalRankStrategy


async def run(total_gb: float, chunk_mb: int, verify: bool,
              capacity_gb: float = None):
    await ts.initialize(
        num_storage_volumes=1,
        strategy=LocalRankStrategy(),
        storage_device="auto",
        storage_capacity_gb=capacity_gb,
    )
    try:
        chunk_bytes = chunk_mb << 20
        n = max(1, int(total_gb * 1e9 / chunk_bytes))
        numel = chunk_bytes // 2
        print(f"payload: {n} x {chunk_mb} MB bf16 chunks "
              f"({n * chunk_bytes / 1e9:.1f} GB)", flush=True)
        items = {}
        for i in range(n):
            t = torch.empty(numel, dtype=torch.bfloat16, device="cuda")
            t.view(torch.int16).fill_(i % 251)
            items[f"cap/{i}"] = t
        torch.cuda.synchronize()

        total = n * chunk_bytes
        t0 = time.perf_counter()
        await ts.put_batch(items)
        torch.cuda.synchronize()
        put_dt = time.perf_counter() - t0
        print(f"put_batch (cold): {put_dt*1e3:.1f} ms  "
              f"{total/put_dt/1e9:.0f} GB/s", flush=True)
        # warm overwrite: exports/opens cached, storage reused in place
        t0 = time.perf_counter()
        await ts.put_batch(items)
        torch.cuda.synchronize()
        put_dt = time.perf_counter() - t0
        print(f"put_batch (warm): {put_dt*1e3:.1f} ms  "
              f"{total/put_dt/1e9:.0f} GB/s", flush=True)
        free, cap = torch.cuda.mem_get_info()
        print(f"client HBM used: {(cap - free)/1e9:.1f} / {cap/1e9:.0f} GB",
              flush=True)

        dests = {k: torch.empty_like(v) for k, v in items.items()}
        t0 = time.perf_counter()
        out = await ts.get_batch(dests)
        torch.cuda.synchronize()
        get_dt = time.perf_counter() - t0
        print(f"get_batch (cold): {get_dt*1e3:.1f} ms  "
              f"{total/get_dt/1e9:.0f} GB/s", flush=True)
        # warm: dest handles already opened by the volume, plan cached
        t0 = time.perf_counter()
        out = await ts.get_batch(dests)
        torch.cuda.synchronize()
        get_dt = time.perf_counter() - t0
        print(f"get_batch (warm): {get_dt*1e3:.1f} ms  "
              f"{total/get_dt/1e9:.0f} GB/s", flush=True)
        if verify:
            for i in (0, n // 2, n - 1):
                k = f"cap/{i}"
                assert torch.equal(out[k], items[k]), k
            print("verify ok", flush=True)
    finally:
        await ts.shutdown()


async def run_stream(total_gb: float, chunk_mb: int, wave_gb: float):
    """Streaming mode: fill the store to ``total_gb`` RESIDENT (the
    288 GB-per-GPU sizing claim) by pushing in waves and freeing the
    client's source copies between waves, then read everything back
    through a reused destination wave with pattern verification."""
    await ts.initialize(
        num_storage_volumes=1,
        strategy=LocalRankStrategy(),
        storage_device="auto",
    )
    try:
        chunk_bytes = chunk_mb << 20
        n = max(1, int(total_gb * 1e9 / chunk_bytes))
        per_wave = max(1, int(wave_gb * 1e9 / chunk_bytes))
        numel = chunk_bytes // 2
        print(f"streaming {n} x {chunk_mb} MB bf16 chunks "
              f"({n * chunk_bytes / 1e9:.1f} GB resident), "
              f"wave={per_wave} chunks", flush=True)
        put_s = 0.0
        for w0 in range(0, n, per_wave):
            wave = {}
            for i in range(w0, min(w0 + per_wave, n)):
                t = torch.empty(numel, dtype=torch.bfloat16, device="cuda")
                t.view(torch.int16).fill_(i % 251)
                wave[f"cap/{i}"] = t
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            await ts.put_batch(wave)
            torch.cuda.synchronize()
            put_s += time.perf_counter() - t0
            wave.clear()
        free, cap = torch.cuda.mem_get_info()
        print(f"put: {n * chunk_bytes / put_s / 1e9:.0f} GB/s sustained; "
              f"device used {(cap - free) / 1e9:.1f} / {cap / 1e9:.0f} GB "
              f"(client + volume store share the GPU)", flush=True)
        get_s = 0.0
        dests = [
            torch.empty(numel, dtype=torch.bfloat16, device="cuda")
            for _ in range(per_wave)
        ]
        bad = 0
        for w0 in range(0, n, per_wave):
            ks = [f"cap/{i}" for i in range(w0, min(w0 + per_wave, n))]
            fetches = dict(zip(ks, dests))
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            await ts.get_batch(fetches)
            torch.cuda.synchronize()
            get_s += time.perf_counter() - t0
            for j, i in enumerate(range(w0, min(w0 + per_wave, n))):
                if not dests[j].view(torch.int16).eq(i % 251).all():
                    bad += 1
        print(f"get: {n * chunk_bytes / get_s / 1e9:.0f} GB/s sustained; "
              f"verify {'ok' if bad == 0 else f'{bad} chunks WRONG'}",
              flush=True)
        assert bad == 0
    finally:
        await ts.shutdown()


def main():
    p = argparse.ArgumentParser()
    # src + stored + dest copies coexist on one 288 GB GPU
    p.add_argument("--total-gb", type=float, default=80.0)
    p.add_argument("--chunk-mb", type=int, default=1024)
    p.add_argument("--no-verify", action="store_true")
    p.add_argument("--stream", action="store_true",
                   help="wave-streamed fill: store holds --total-gb "
                        "resident (client frees sources between waves)")
    p.add_argument("--wave-gb", type=float, default=16.0)
    p.add_argument("--capacity-gb", type=float, default=None,
                   help="TieredStore primary capacity (overflow spills to host)")
    args = p.parse_args()
    if args.stream:
        asyncio.run(run_stream(args.total_gb, args.chunk_mb, args.wave_gb))
    else:
        asyncio.run(run(args.total_gb, args.chunk_mb, not args.no_verify,
                        args.capacity_gb))


if __name__ == "__main__":
    main()
