"""gfx950 kernel numerics vs plain PyTorch fp32 references (the HIP path is
the one under test — these fail loudly if the extension did not load)."""
import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs an AMD GPU")


@requires_gpu
@pytest.mark.parametrize("dtype", [torch.float32, torch.float64, torch.bfloat16,
                                   torch.float16, torch.int32, torch.int64])
@pytest.mark.parametrize("n", [1, 17, 4096, 1 << 20, (1 << 20) + 5])
def test_reduce_into_sum(dtype, n):
    from torchmpi_amd.ops import reduce_into
    torch.manual_seed(0)
    if dtype.is_floating_point:
        a = torch.randn(n, device="cuda").to(dtype)
        b = torch.randn(n, device="cuda").to(dtype)
    else:
        a = torch.randint(-1000, 1000, (n,), device="cuda", dtype=dtype)
        b = torch.randint(-1000, 1000, (n,), device="cuda", dtype=dtype)
    if dtype in (torch.bfloat16, torch.float16):
        ref = a.float() + b.float()  # kernel accumulates in fp32
    else:
        ref = a + b
    reduce_into(a, b, "sum")
    torch.cuda.synchronize()
    if dtype in (torch.bfloat16, torch.float16):
        # one rounding of the fp32 sum back to bf16/f16
        assert torch.allclose(a.float(), ref, rtol=1e-2, atol=1e-2)
    else:
        assert torch.equal(a, ref)


@requires_gpu
@pytest.mark.parametrize("op,fn", [("max", torch.maximum), ("min", torch.minimum)])
def test_reduce_into_minmax(op, fn):
    from torchmpi_amd.ops import reduce_into
    a = torch.randn(100000, device="cuda")
    b = torch.randn(100000, device="cuda")
    ref = fn(a, b)
    reduce_into(a, b, op)
    torch.cuda.synchronize()
    assert torch.equal(a, ref)


@requires_gpu
def test_reduce_unaligned_offsets():
    """Ring segments can start at any element offset; the kernel must handle
    non-16B-aligned pointers (scalar fallback path)."""
    from torchmpi_amd.ops import reduce_into
    base_a = torch.randn(1001, device="cuda", dtype=torch.bfloat16)
    base_b = torch.randn(1001, device="cuda", dtype=torch.bfloat16)
    a, b = base_a[3:], base_b[3:]
    ref = a.float() + b.float()
    reduce_into(a.contiguous(), b.contiguous(), "sum")  # contiguous slices
    # direct in-place on a view with odd offset:
    av, bv = base_a[1:], base_b[1:]
    ref2 = av.float() + bv.float()
    from torchmpi_amd import _C
    _C.reduce_into(av, bv, 0)
    torch.cuda.synchronize()
    assert torch.allclose(av.float(), ref2, rtol=1e-2, atol=1e-2)


@requires_gpu
def test_scale():
    from torchmpi_amd.ops import scale_
    for dtype in (torch.float32, torch.bfloat16):
        t = torch.randn(123457, device="cuda").to(dtype)
        ref = (t.float() * 0.125).to(dtype)
        scale_(t, 0.125)
        torch.cuda.synchronize()
        assert torch.allclose(t.float(), ref.float(), rtol=1e-3, atol=1e-3)


@requires_gpu
@pytest.mark.parametrize("bucket_dtype,tensor_dtype", [
    (torch.float32, torch.float32),
    (torch.bfloat16, torch.bfloat16),
    (torch.bfloat16, torch.float32),  # cast-fused compression
])
def test_bucket_pack_unpack(bucket_dtype, tensor_dtype):
    from torchmpi_amd.ops import Bucket
    torch.manual_seed(1)
    ts = [torch.randn(n, device="cuda").to(tensor_dtype)
          for n in (5, 1000, 333, 1 << 16)]
    orig = [t.clone() for t in ts]
    b = Bucket(ts, bucket_dtype=bucket_dtype)
    b.pack()
    torch.cuda.synchronize()
    # packed values equal casts of the originals
    for t, off, o in zip(ts, b.offsets, orig):
        seg = b.buffer[off:off + t.numel()]
        assert torch.allclose(seg.float(), o.to(bucket_dtype).float())
    for t in ts:
        t.zero_()
    b.unpack(alpha=2.0)
    torch.cuda.synchronize()
    tol = 1e-2 if bucket_dtype == torch.bfloat16 else 1e-6
    for t, o in zip(ts, orig):
        assert torch.allclose(t.float(), 2.0 * o.to(bucket_dtype).float(),
                              rtol=tol, atol=tol)


@requires_gpu
def test_bucket_hole_zeroing():
    """Pad holes between tensors must stay zero after pack (bf16 overflow
    safety for the allreduce)."""
    from torchmpi_amd.ops import Bucket
    ts = [torch.ones(3, device="cuda"), torch.ones(5, device="cuda")]
    b = Bucket(ts)
    b.pack()
    torch.cuda.synchronize()
    assert b.offsets[1] == 4  # 3 rounded up to 16B/4B = 4 elements
    assert b.buffer[3].item() == 0.0
    assert b.buffer[b.offsets[1] + 5:].abs().sum().item() == 0.0


@requires_gpu
def test_native_extension_is_loaded():
    """Guard against silent eager fallback: the in-tree _C extension must be
    the loaded module."""
    import torchmpi_amd
    path = torchmpi_amd._C.__file__
    assert "torchmpi_amd" in path and path.endswith(".so"), path
