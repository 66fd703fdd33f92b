"""Tile/grid tuning sweep for the copy_slices kernel (run on MI355X).

HIPSTORE_TILE / HIPSTORE_GRID / HIPSTORE_NT are read per call, so one
process sweeps every combination.  Patterns cover the three hot shapes:
flat bulk copy, strided pack (reshard extract), small-row scatter
(reshard assembly), plus the K3 cast.

    python benchmarks/kernel_tune.py > gpurun_out/kernel_tune.log
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from torchstore_amd.ops import gpu


def timeit(fn, n=10):
    fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n


def main():
    torch.cuda.set_device(0)
    e = gpu.ext()
    dev = torch.device("cuda", 0)

    # patterns
    n1g = 1 << 30
    a1 = torch.empty(n1g, dtype=torch.uint8, device=dev)
    b1 = torch.empty(n1g, dtype=torch.uint8, device=dev)

    big = torch.randn(8192, 8192, device=dev)
    view = big[:, 2048:6144]
    packed = torch.empty(view.shape, dtype=big.dtype, device=dev)

    n_slices = 256
    srcs = [torch.randn(512, 512, device=dev, dtype=torch.bfloat16)
            for _ in range(n_slices)]
    dest_big = torch.zeros(512, 512 * n_slices, device=dev,
                           dtype=torch.bfloat16)
    pairs = [(srcs[i], dest_big[:, i * 512: (i + 1) * 512])
             for i in range(n_slices)]

    f32 = torch.randn(1 << 28, device=dev)
    bf = torch.empty(1 << 28, dtype=torch.bfloat16, device=dev)

    def flat():
        e.copy_batch([(b1.data_ptr(), 0, a1.data_ptr(), 0, n1g)])

    def pack():
        gpu.copy_pairs([(view, packed)], dev, blocking=True)

    def scatter():
        gpu.copy_pairs(pairs, dev, blocking=True)

    def cast():
        gpu.cast_copy(f32, bf)

    bytes_moved = {
        "flat": 2 * n1g,
        "pack": view.numel() * 8,
        "scatter": 2 * n_slices * 512 * 512 * 2,
        "cast": f32.numel() * 6,
    }
    fns = {"flat": flat, "pack": pack, "scatter": scatter, "cast": cast}

    print(f"{'pattern':8s} {'tile':>7s} {'grid':>5s} {'nt':>2s} "
          f"{'ms':>8s} {'GB/s':>8s}")
    best = {}
    for name, fn in fns.items():
        for nt in ("1", "0"):
            os.environ["HIPSTORE_NT"] = nt
            for tile in (16384, 32768, 65536, 131072, 262144):
                os.environ["HIPSTORE_TILE"] = str(tile)
                for grid in (1024, 2048, 4096, 8192):
                    os.environ["HIPSTORE_GRID"] = str(grid)
                    dt = timeit(fn)
                    bw = bytes_moved[name] / dt / 1e9
                    print(f"{name:8s} {tile:7d} {grid:5d} {nt:>2s} "
                          f"{dt * 1e3:8.3f} {bw:8.1f}", flush=True)
                    if bw > best.get(name, (0, None))[0]:
                        best[name] = (bw, (tile, grid, nt))
    for k in ("HIPSTORE_NT", "HIPSTORE_TILE", "HIPSTORE_GRID"):
        os.environ.pop(k, None)
    print("\nbest:")
    for name, (bw, cfg) in best.items():
        print(f"  {name}: {bw:.1f} GB/s at tile={cfg[0]} grid={cfg[1]} "
              f"nt={cfg[2]}")


if __name__ == "__main__":
    main()
