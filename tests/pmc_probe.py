"""Minimal kernel-only workload for rocprofv3 --pmc (no randn, no mp)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from torchstore_amd.ops import gpu


def main():
    torch.cuda.set_device(0)
    big = torch.empty(8192, 8192, device="cuda")
    view = big[:, 2048:6144]
    out = torch.empty(view.shape, dtype=big.dtype, device="cuda")
    src = torch.empty(1 << 28, device="cuda")
    dst = torch.empty(1 << 28, dtype=torch.bfloat16, device="cuda")
    for _ in range(5):
        gpu.ext().copy_slices(
            [(view.data_ptr(), out.data_ptr(), 4096 * 4, [8192], [8192 * 4],
              [4096 * 4])],
            0, torch.cuda.current_stream().cuda_stream, True,
        )
        gpu.cast_copy(src, dst)
    torch.cuda.synchronize()
    print("pmc probe done")


if __name__ == "__main__":
    main()
