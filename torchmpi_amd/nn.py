"""Model synchronization layer (reference: torchmpi/nn.lua, SURVEY.md
sections 2.2 and 3.3/3.5).

  synchronize_parameters(model)  broadcast parameters from root so replicas
                                 agree (reference mpinn.synchronizeParameters)
  synchronize_gradients(model)   sum-allreduce all gradients (reference
                                 mpinn.synchronizeGradients), bucket-fused
  GradientSynchronizer           the async overlap engine: bucketed gradient
                                 allreduce launched from backward hooks on
                                 dedicated comm streams, overlapping the rest
                                 of backward (reference stack 3.3)
"""
from typing import List, Optional

import torch

from .ops import Bucket
from .utils import env_int, get_logger


def _params(model) -> List[torch.nn.Parameter]:
    return [p for p in model.parameters() if p.requires_grad]


def synchronize_parameters(model: torch.nn.Module, root: int = 0, comm=None):
    """Broadcast every parameter (and buffer) from `root`."""
    from . import broadcast, communicator
    c = comm or communicator()
    if c.size() == 1:
        return
    with torch.no_grad():
        for t in list(model.parameters()) + list(model.buffers()):
            if t.is_floating_point() or t.dtype in (torch.int64, torch.int32):
                broadcast(t.data if hasattr(t, "data") else t, root=root, comm=c)


def synchronize_gradients(model: torch.nn.Module, average: bool = True, comm=None):
    """One-shot bucketed gradient allreduce (sync)."""
    from . import communicator
    c = comm or communicator()
    if c.size() == 1:
        return
    sync = GradientSynchronizer(model, comm=c, average=average, _register_hooks=False)
    sync.reduce_all_sync()


class GradientSynchronizer:
    """Bucketed backward-overlap gradient allreduce (DDP-style, built on the
    torchmpi_amd engine; re-designs the reference's async flat-view gradient
    allreduce, SURVEY.md stack 3.3).

    Usage::

        sync = GradientSynchronizer(model, bucket_mb=25)
        for step:
            loss.backward()     # buckets fire as their grads finalize
            sync.finish()       # flush + wait + unpack (stream-ordered)
            optimizer.step()
            optimizer.zero_grad(set_to_none=False)   # REQUIRED

    Grad tensors are pre-allocated and must stay in place (zero_grad with
    set_to_none=False), so the bucket meta tables are built once.
    comm_dtype=torch.bfloat16 with fp32 params gives cast-fused 2x smaller
    buckets (the reference's flat-view + our pack kernel's cast fusion).
    """

    def __init__(self, model: torch.nn.Module, comm=None, bucket_mb: int = 0,
                 average: bool = True, comm_dtype: Optional[torch.dtype] = None,
                 _register_hooks: bool = True):
        from . import communicator
        self.comm = comm or communicator()
        self.model = model
        self.average = average
        self.log = get_logger()
        bucket_mb = bucket_mb or env_int("TORCHMPI_BUCKET_MB", 25)
        self.bucket_bytes = bucket_mb << 20

        params = _params(model)
        # Reverse registration order approximates backward completion order.
        params = list(reversed(params))
        # Pre-allocate grads so pointers are stable.
        for p in params:
            if p.grad is None:
                p.grad = torch.zeros_like(p)

        # Greedy fixed bucket assignment by dtype.
        self.buckets = []
        cur, cur_bytes, cur_dtype = [], 0, None
        for p in params:
            pb = p.numel() * p.element_size()
            if cur and (cur_dtype != p.dtype or cur_bytes + pb > self.bucket_bytes):
                self.buckets.append(self._make_bucket(cur, comm_dtype))
                cur, cur_bytes = [], 0
            cur.append(p)
            cur_bytes += pb
            cur_dtype = p.dtype
        if cur:
            self.buckets.append(self._make_bucket(cur, comm_dtype))

        self._param_bucket = {}
        for bi, b in enumerate(self.buckets):
            for p in b["params"]:
                self._param_bucket[id(p)] = bi

        self._hooks = []
        self._accum_mode = False
        if _register_hooks and self.comm.size() > 1:
            for p in params:
                h = p.register_post_accumulate_grad_hook(self._hook)
                self._hooks.append(h)
        self._reset_state()

    def _make_bucket(self, params, comm_dtype):
        grads = [p.grad for p in params]
        bucket = Bucket(grads, bucket_dtype=comm_dtype or params[0].dtype)
        return {"params": params, "bucket": bucket, "pending": 0, "handle": None}

    def _reset_state(self):
        for b in self.buckets:
            b["pending"] = len(b["params"])
            b["handle"] = None
            b["ready"] = False
        self._next_launch = 0

    # -- hot path ----------------------------------------------------------
    # Buckets LAUNCH in fixed bucket-index order even if they become ready
    # out of order: RCCL grouped-p2p ops match by per-communicator call
    # order, so every rank must issue its bucket allreduces in the same
    # sequence (the reference's per-communicator thread gave the same
    # guarantee; DDP enforces the same rule).

    def _hook(self, p):
        if self._accum_mode:
            return  # no_sync(): grads accumulate locally, no comm
        bi = self._param_bucket[id(p)]
        b = self.buckets[bi]
        if b["pending"] <= 0:
            raise RuntimeError(
                "GradientSynchronizer: a gradient hook fired on an "
                "already-reduced bucket — multiple backward() calls before "
                "finish() would silently overwrite the accumulated grads "
                "with a stale reduction. Wrap the non-final backward passes "
                "in sync.no_sync() for gradient accumulation.")
        b["pending"] -= 1
        if b["pending"] == 0:
            b["ready"] = True
            self._launch_ready_in_order()

    def no_sync(self):
        """Context manager: skip gradient allreduce inside (DDP-style
        gradient accumulation). The final backward runs outside no_sync()
        and reduces the accumulated grads."""
        import contextlib

        @contextlib.contextmanager
        def ctx():
            prev = self._accum_mode
            self._accum_mode = True
            try:
                yield
            finally:
                self._accum_mode = prev
        return ctx()

    def _launch_ready_in_order(self):
        while (self._next_launch < len(self.buckets)
               and self.buckets[self._next_launch]["ready"]):
            self._launch(self.buckets[self._next_launch])
            self._next_launch += 1

    def _launch(self, b):
        from . import allreduce
        bucket = b["bucket"]
        if b["params"][0].grad is not bucket.tensors[0]:
            raise RuntimeError(
                "GradientSynchronizer: grad tensors were reallocated — use "
                "optimizer.zero_grad(set_to_none=False) so the registered "
                "bucket buffers stay valid")
        bucket.pack()  # on current stream (autograd stream)
        b["handle"] = allreduce(bucket.buffer, async_op=True, comm=self.comm)

    def finish(self):
        """Flush un-launched buckets (fixed order), wait for all allreduces
        (stream-level), unpack with the 1/W average fused."""
        if self.comm.size() == 1:
            return
        alpha = 1.0 / self.comm.size() if self.average else 1.0
        for b in self.buckets:
            if b["handle"] is None:
                self._launch(b)  # unused params / not-yet-ready buckets
        for b in self.buckets:
            b["handle"].wait()  # current stream waits on comm streams
            b["bucket"].unpack(alpha)
        self._reset_state()

    def reduce_all_sync(self):
        """Non-overlapped path: pack + allreduce + unpack every bucket."""
        alpha = 1.0 / self.comm.size() if self.average else 1.0
        for b in self.buckets:
            self._launch(b)
        for b in self.buckets:
            b["handle"].wait()
            b["bucket"].unpack(alpha)
        self._reset_state()

    def detach(self):
        for h in self._hooks:
            h.remove()
        self._hooks = []
