"""GPU tensor ops backed by the gfx950 kernels (csrc/kernels/*.hip).

These are the hand-written CDNA4 pieces of the collective engine, exposed
for direct use and for numerics tests:
  - reduce_into / scale_: the ring-step local reduction (reference THC axpy,
    SURVEY.md section 2.4 row 7)
  - Bucket: multi-tensor pack/unpack with cast + average fusion (reference
    flat-view fusion, row 8)

They fail loudly if the native extension is missing — there is no eager
fallback on GPU.
"""
from typing import List, Optional

import torch

from .. import _C

_DTYPE_CODE = {
    torch.float32: 0,
    torch.float64: 1,
    torch.float16: 2,
    torch.bfloat16: 3,
    torch.int32: 4,
    torch.int64: 5,
    torch.uint8: 6,
    torch.int8: 7,
}


def reduce_into(dst: torch.Tensor, src: torch.Tensor, op="sum") -> torch.Tensor:
    """dst = dst (op) src elementwise, on the current stream (gfx950 kernel).
    bf16/f16 accumulate in fp32 in-register."""
    from .. import _OPS
    opv = _OPS[op] if isinstance(op, str) else op
    _C.reduce_into(dst, src, int(opv))
    return dst


def scale_(t: torch.Tensor, alpha: float) -> torch.Tensor:
    _C.scale_(t, float(alpha))
    return t


def _align_elems(dtype: torch.dtype) -> int:
    return 16 // torch.tensor([], dtype=dtype).element_size()


class Bucket:
    """A fixed set of tensors fused into one flat communication buffer.

    Offsets are 16-byte aligned (the pack kernels' vectorization contract);
    the bucket buffer is allocated once and the device-side meta table is
    built once. pack()/unpack() are single kernel launches.
    """

    def __init__(self, tensors: List[torch.Tensor],
                 bucket_dtype: Optional[torch.dtype] = None,
                 device: Optional[torch.device] = None):
        assert tensors, "empty bucket"
        t0 = tensors[0]
        self.tensors = tensors
        self.tensor_dtype = t0.dtype
        assert all(t.dtype == self.tensor_dtype for t in tensors), \
            "bucket tensors must share a dtype"
        self.bucket_dtype = bucket_dtype or self.tensor_dtype
        device = device or t0.device
        align = _align_elems(self.bucket_dtype)
        offsets = []
        off = 0
        for t in tensors:
            offsets.append(off)
            off += (t.numel() + align - 1) // align * align
        self.total = off
        self.offsets = offsets
        self.buffer = torch.zeros(self.total, dtype=self.bucket_dtype, device=device)
        if device.type == "cuda":
            self._meta = _C.build_pack_meta(list(tensors), offsets)
        else:
            self._meta = None

    def pack(self) -> torch.Tensor:
        if self._meta is not None:
            _C.pack_bucket(self.buffer, self._meta, len(self.tensors),
                           _DTYPE_CODE[self.tensor_dtype])
        else:  # CPU reference path (tests)
            for t, off in zip(self.tensors, self.offsets):
                self.buffer[off:off + t.numel()] = t.reshape(-1).to(self.bucket_dtype)
        return self.buffer

    def unpack(self, alpha: float = 1.0) -> None:
        if self._meta is not None:
            _C.unpack_bucket(self.buffer, self._meta, len(self.tensors),
                             _DTYPE_CODE[self.tensor_dtype], float(alpha))
        else:
            for t, off in zip(self.tensors, self.offsets):
                seg = self.buffer[off:off + t.numel()].to(t.dtype)
                if alpha != 1.0:
                    seg = seg * alpha
                t.reshape(-1).copy_(seg)


__all__ = ["reduce_into", "scale_", "Bucket"]
