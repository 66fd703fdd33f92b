"""Numerics of the CDNA4 HIP kernels vs plain PyTorch fp32 references."""

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(
    not torch.cuda.is_available(), reason="needs an AMD GPU"
)


@requires_gpu
def test_extension_loads_natively():
    from torchstore_amd.ops import gpu

    e = gpu.ext()
    assert e.device_count() >= 1


@requires_gpu
@pytest.mark.parametrize(
    "src_dtype,dst_dtype",
    [
        (torch.float32, torch.bfloat16),
        (torch.bfloat16, torch.float32),
        (torch.float32, torch.float16),
        (torch.float16, torch.float32),
    ],
)
def test_cast_copy_matches_torch(src_dtype, dst_dtype):
    from torchstore_amd.ops import gpu

    for numel in [1, 7, 256, 4096 + 3, 1 << 20]:
        src = torch.randn(numel, dtype=torch.float32, device="cuda").to(src_dtype)
        out = torch.empty(numel, dtype=dst_dtype, device="cuda")
        gpu.cast_copy(src, out)
        ref = src.to(dst_dtype)  # torch eager reference
        torch.cuda.synchronize()
        assert torch.equal(out, ref), f"mismatch at numel={numel}"


@requires_gpu
def test_copy_pairs_gather():
    """K1: strided region -> contiguous pack equals torch slicing."""
    from torchstore_amd.ops import gpu

    t = torch.randn(128, 256, device="cuda")
    view = t[17:93, 31:200]
    out = torch.empty(view.shape, dtype=t.dtype, device="cuda")
    gpu.copy_pairs([(view, out)], t.device)
    torch.cuda.synchronize()
    assert torch.equal(out, view.contiguous())


@requires_gpu
def test_copy_pairs_scatter():
    """K2: contiguous parts -> strided destination regions."""
    from torchstore_amd.ops import gpu

    dest = torch.zeros(64, 64, device="cuda")
    ref = torch.zeros(64, 64, device="cuda")
    parts = []
    for r0, r1, c0, c1 in [(0, 32, 0, 64), (32, 64, 0, 32), (32, 64, 32, 64)]:
        src = torch.randn(r1 - r0, c1 - c0, device="cuda")
        parts.append((src, dest[r0:r1, c0:c1]))
        ref[r0:r1, c0:c1] = src
    gpu.copy_pairs(parts, dest.device)
    torch.cuda.synchronize()
    assert torch.equal(dest, ref)


@requires_gpu
def test_copy_pairs_3d_and_bf16():
    from torchstore_amd.ops import gpu

    t = torch.randn(8, 64, 96, device="cuda").to(torch.bfloat16)
    view = t[2:7, 5:60, 11:80]
    out = torch.empty(view.shape, dtype=t.dtype, device="cuda")
    gpu.copy_pairs([(view, out)], t.device)
    torch.cuda.synchronize()
    assert torch.equal(out, view.contiguous())


@requires_gpu
def test_copy_pairs_odd_alignment():
    """bf16 with odd column offsets exercises the sub-16B paths."""
    from torchstore_amd.ops import gpu

    t = torch.randn(33, 131, device="cuda").to(torch.bfloat16)
    view = t[1:32, 3:128]
    out = torch.empty(view.shape, dtype=t.dtype, device="cuda")
    gpu.copy_pairs([(view, out)], t.device)
    torch.cuda.synchronize()
    assert torch.equal(out, view.contiguous())


@requires_gpu
def test_pack_region_large():
    from torchstore_amd.ops import gpu

    t = torch.randn(4096, 4096, device="cuda")
    view = t[:, 1024:3072]
    out = gpu.pack_region(view)
    torch.cuda.synchronize()
    assert torch.equal(out, view.contiguous())


@requires_gpu
def test_cast_tensor_api():
    from torchstore_amd.ops.cast import cast_tensor

    f = torch.randn(1 << 16, device="cuda")
    b = cast_tensor(f, torch.bfloat16)
    torch.cuda.synchronize()
    assert torch.equal(b, f.to(torch.bfloat16))


@requires_gpu
def test_copy_pairs_row_packed_boundaries():
    """Row-packed units (small aligned rows): row counts chosen to hit
    partial last units and slice transitions inside one wave's chunk."""
    import torch

    from torchstore_amd.ops import gpu

    torch.cuda.set_device(0)
    pairs = []
    expect = []
    for rows, cols in [(3, 8), (17, 64), (511, 128), (1000, 8), (16, 2048)]:
        big = torch.randn(rows, cols * 3, device="cuda")
        src = big[:, cols : 2 * cols]  # strided, 16B-aligned rows (f32)
        dst = torch.zeros(rows, cols, device="cuda")
        pairs.append((src, dst))
        expect.append(src.clone())
    gpu.copy_pairs(pairs, torch.device("cuda", 0), blocking=True)
    torch.cuda.synchronize()
    for (src, dst), exp in zip(pairs, expect):
        assert torch.equal(dst, exp)
