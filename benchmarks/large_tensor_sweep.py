"""Large-tensor put/get sweep (the reference's benchmark harness shape:
tests/test_large_tensors.py writes put/get CSVs of size_mbytes, delta, MB/s).

Usage:
    python benchmarks/large_tensor_sweep.py --sizes-mb 4 64 1024 2048 \
        --device cuda --csv profiles/large_tensor_sweep.csv
"""

from __future__ import annotations

import argparse
import asyncio
import csv
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import torchstore_amd as ts
from torchstore_amd.strategy import SingletonStrategy
from torchstore_amd.transport import TransportType


async def sweep(sizes_mb, device, transport, csv_path, repeats=3):
    await ts.initialize(
        num_storage_volumes=1,
        strategy=SingletonStrategy(
            transport=TransportType(transport) if transport else None
        ),
        storage_device="auto" if device == "cuda" else "cpu",
    )
    rows = []
    try:
        for mb in sizes_mb:
            numel = mb * (1 << 20) // 4
            t = torch.randn(numel, dtype=torch.float32, device=device)
            dest = torch.empty_like(t)
            # warm (allocates segments / opens handles)
            await ts.put("sweep", t)
            await ts.get("sweep", dest)
            put_dt = get_dt = 0.0
            for _ in range(repeats):
                if device == "cuda":
                    torch.cuda.synchronize()
                t0 = time.perf_counter()
                await ts.put("sweep", t)
                put_dt += time.perf_counter() - t0
                t0 = time.perf_counter()
                await ts.get("sweep", dest)
                get_dt += time.perf_counter() - t0
            put_dt /= repeats
            get_dt /= repeats
            assert torch.equal(dest, t), f"roundtrip mismatch at {mb}MB"
            rows.append(
                {
                    "size_mbytes": mb,
                    "put_s": round(put_dt, 6),
                    "put_MBps": round(mb / put_dt, 1),
                    "get_s": round(get_dt, 6),
                    "get_MBps": round(mb / get_dt, 1),
                    "device": device,
                    "transport": transport or "auto",
                }
            )
            print(rows[-1], flush=True)
            await ts.delete("sweep")
            del t, dest
    finally:
        await ts.shutdown()
    if csv_path:
        os.makedirs(os.path.dirname(csv_path) or ".", exist_ok=True)
        with open(csv_path, "w", newline="") as f:
            w = csv.DictWriter(f, fieldnames=list(rows[0].keys()))
            w.writeheader()
            w.writerows(rows)
    return rows


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--sizes-mb", type=int, nargs="+",
                   default=[4, 16, 64, 256, 1024, 2048])
    p.add_argument("--device", default="cuda" if torch.cuda.is_available() else "cpu")
    p.add_argument("--transport", default=None)
    p.add_argument("--csv", default=None)
    p.add_argument("--repeats", type=int, default=3)
    args = p.parse_args()
    asyncio.run(
        sweep(args.sizes_mb, args.device, args.transport, args.csv, args.repeats)
    )


if __name__ == "__main__":
    main()
