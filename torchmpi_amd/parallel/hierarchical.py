"""Two-level hierarchical allreduce.

Reference: TorchMPI's hierarchical intra-node (NCCL/IPC) + inter-node (MPI)
paths in lib/collectives_cuda.cpp. Here both
levels ride the same engine type; the composition is:

    intra.reduce_scatter -> inter.allreduce(shard) -> intra.allgather

which moves 2(Wi-1)/Wi + 2(We-1)/(Wi*We) ... of the data per level instead
of running the flat ring twice.
"""
import torch


def hierarchical_allreduce(tensor: torch.Tensor, intra, inter, average=False):
    """Allreduce `tensor` over the product group intra x inter.

    `intra` and `inter` are Communicators from world.split(): every rank is
    in exactly one intra group and one inter group, and inter groups connect
    ranks holding the same shard index.
    """
    from .. import allgather, allreduce, reduce_scatter_

    wi = intra.size()
    n = tensor.numel()
    base, rem = divmod(n, wi)
    my_count = base + (1 if intra.rank() < rem else 0)
    flat = tensor.view(-1)
    shard = torch.empty(my_count, dtype=tensor.dtype, device=tensor.device)

    # In-place variant: flat is rewritten by the allgather below anyway,
    # so the copy the preserving reduce_scatter makes would be wasted.
    reduce_scatter_(flat, shard, comm=intra)
    # The value is summed over all wi*we ranks, so the mean divides by the
    # product — averaging only at the inter level would leave it wi x too big.
    allreduce(shard, comm=inter)
    if average:
        shard /= float(wi * inter.size())

    if rem == 0:
        # equal shards: allgather straight back into the tensor
        out = torch.empty_like(flat)
        allgather(shard, out=out, comm=intra)
        flat.copy_(out.view(-1))
    else:
        # unequal shards: pad to the max shard size for the allgather
        pad = torch.zeros(base + 1, dtype=tensor.dtype, device=tensor.device)
        pad[:my_count] = shard
        out = torch.empty(wi * (base + 1), dtype=tensor.dtype, device=tensor.device)
        allgather(pad, out=out, comm=intra)
        pieces = []
        for r in range(wi):
            c = base + (1 if r < rem else 0)
            pieces.append(out[r * (base + 1):r * (base + 1) + c])
        flat.copy_(torch.cat(pieces))
    return tensor
