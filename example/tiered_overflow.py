"""TieredStore demo: HBM-primary volumes with host-memory overflow.

When the working set outgrows the primary capacity, later puts spill to
pinned host memory transparently — reads serve from either tier.

Run (CPU demo, any machine; on a GPU node the primary tier is HBM):
    python example/tiered_overflow.py
"""

import asyncio
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import torchstore_amd as ts


async def main():
    await ts.initialize(
        num_storage_volumes=1,
        strategy=ts.SingletonStrategy(),
        storage_device="auto",
        storage_capacity_gb=0.001,  # 1 MB primary for the demo
    )
    try:
        small = torch.randn(64, 64)          # 16 kB -> primary tier
        big = torch.randn(512, 1024)         # 2 MB  -> spills
        await ts.put_batch({"model/head": small, "model/body": big})

        stats = await ts.stats()
        v = stats["volumes"][0]
        print(f"volume {v['volume_id']}: {v['entries']} entries, "
              f"{v['tensor_bytes']} tensor bytes, "
              f"primary {v['tier_primary_used']}/{v['tier_capacity']} B")

        for k, src in (("model/head", small), ("model/body", big)):
            out = await ts.get(k)
            assert torch.equal(out.cpu(), src), k
            print(f"{k}: read back ok ({out.numel() * out.element_size()} B)")
    finally:
        await ts.shutdown()


if __name__ == "__main__":
    asyncio.run(main())
