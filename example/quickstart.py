"""Quickstart: put/get tensors and objects through the store.

Run:  python example/quickstart.py
"""

import asyncio
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import torchstore_amd as ts


async def main():
    # one storage volume (GPU-resident when a GPU is visible) + controller
    await ts.initialize(num_storage_volumes=1)

    device = "cuda" if torch.cuda.is_available() else "cpu"
    w = torch.randn(1024, 1024, device=device)

    await ts.put("model/w", w)
    await ts.put("model/config", {"layers": 12, "heads": 8})

    out = await ts.get("model/w")
    assert torch.equal(out, w)
    cfg = await ts.get("model/config")
    print("config:", cfg)

    # in-place get into a preallocated destination
    dest = torch.zeros_like(w)
    await ts.get("model/w", dest)
    assert torch.equal(dest, w)

    # slicing the keyspace
    print("keys under model/:", sorted(await ts.keys("model")))
    await ts.delete("model/config")
    print("exists after delete:", await ts.exists("model/config"))

    await ts.shutdown()
    print("ok")


if __name__ == "__main__":
    asyncio.run(main())
