"""TP/SP collective algebra as autograd functions
(reference: colossalai/shardformer/layer/_operation.py).

MI355X notes: collectives ride RCCL over the all-to-all xGMI fabric. The
column-parallel linear backward overlaps the input-grad all-reduce with the
weight-grad GEMM (async_op + hipBLASLt on the compute stream) — with
HIP_DEVICE_MAX_CONNECTIONS=1 the reduce is enqueued first and the GEMM
fills the bubble, same discipline as the reference's Megatron-style path.
"""

from typing import Optional

import torch
import torch.distributed as dist

__all__ = [
    "linear_with_async_comm",
    "reduce_forward",
    "reduce_backward",
    "gather_forward_split_backward",
    "split_forward_gather_backward",
    "all_to_all_comm",
    "gather_forward_reduce_scatter_backward",
    "reduce_scatter_forward_gather_backward",
]


def _all_reduce(x: torch.Tensor, group, async_op: bool = False):
    if group is None or dist.get_world_size(group) == 1:
        return x, None
    work = dist.all_reduce(x, group=group, async_op=async_op)
    return x, work


def _split(x: torch.Tensor, dim: int, group) -> torch.Tensor:
    world = dist.get_world_size(group)
    if world == 1:
        return x
    rank = dist.get_rank(group)
    assert x.size(dim) % world == 0, f"dim {dim} size {x.size(dim)} not divisible by tp size {world}"
    chunk = x.size(dim) // world
    return x.narrow(dim, rank * chunk, chunk).contiguous()


def _gather(x: torch.Tensor, dim: int, group) -> torch.Tensor:
    world = dist.get_world_size(group)
    if world == 1:
        return x
    x = x.contiguous()
    if dim == 0:
        out = torch.empty((x.size(0) * world,) + tuple(x.shape[1:]), dtype=x.dtype, device=x.device)
        dist.all_gather_into_tensor(out, x, group=group)
        return out
    parts = [torch.empty_like(x) for _ in range(world)]
    dist.all_gather(parts, x, group=group)
    return torch.cat(parts, dim=dim)


def _reduce_scatter(x: torch.Tensor, dim: int, group) -> torch.Tensor:
    world = dist.get_world_size(group)
    if world == 1:
        return x
    if dim != 0:
        # move dim to front for reduce_scatter_tensor, then back
        x = x.movedim(dim, 0).contiguous()
        out = torch.empty((x.size(0) // world,) + tuple(x.shape[1:]), dtype=x.dtype, device=x.device)
        dist.reduce_scatter_tensor(out, x, group=group)
        return out.movedim(0, dim).contiguous()
    x = x.contiguous()
    out = torch.empty((x.size(0) // world,) + tuple(x.shape[1:]), dtype=x.dtype, device=x.device)
    dist.reduce_scatter_tensor(out, x, group=group)
    return out


def _all_to_all(x: torch.Tensor, scatter_dim: int, gather_dim: int, group) -> torch.Tensor:
    world = dist.get_world_size(group)
    if world == 1:
        return x
    if dist.get_backend(group) == "gloo":
        # gloo has no alltoall: emulate with all_gather (CPU tests only)
        rank = dist.get_rank(group)
        gathered = [torch.empty_like(x) for _ in range(world)]
        dist.all_gather(gathered, x.contiguous(), group=group)
        parts = [g.chunk(world, dim=scatter_dim)[rank] for g in gathered]
        return torch.cat(parts, dim=gather_dim)
    inputs = [t.contiguous() for t in x.chunk(world, dim=scatter_dim)]
    outputs = [torch.empty_like(inputs[0]) for _ in range(world)]
    dist.all_to_all(outputs, inputs, group=group)
    return torch.cat(outputs, dim=gather_dim)


class LinearWithAsyncCommunication(torch.autograd.Function):
    """y = x @ W^T (+ b); column-parallel linear body.

    Backward: dX = dY @ W needs an all-reduce over the tp group; it is issued
    async and overlapped with the dW / db GEMMs.
    """

    @staticmethod
    def forward(ctx, x, weight, bias, group, async_grad_allreduce):
        ctx.save_for_backward(x, weight)
        ctx.use_bias = bias is not None
        ctx.group = group
        ctx.async_grad_allreduce = async_grad_allreduce
        out = torch.nn.functional.linear(x, weight, bias)
        return out

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1])
        x2 = x.reshape(-1, x.shape[-1])
        dx = dy @ weight  # [.., in]
        work = None
        if ctx.async_grad_allreduce:
            _, work = _all_reduce(dx, ctx.group, async_op=True)
        dw = dy2.t() @ x2
        db = dy2.sum(0) if ctx.use_bias else None
        if work is not None:
            work.wait()
        return dx, dw, db, None, None


def linear_with_async_comm(x, weight, bias, group, async_grad_allreduce=True):
    return LinearWithAsyncCommunication.apply(x, weight, bias, group, async_grad_allreduce)


class _ReduceForward(torch.autograd.Function):
    """all-reduce in forward (row-parallel output), identity backward."""

    @staticmethod
    def forward(ctx, x, group):
        x, _ = _all_reduce(x, group)
        return x

    @staticmethod
    def backward(ctx, dy):
        return dy, None


def reduce_forward(x, group):
    return _ReduceForward.apply(x, group)


class _ReduceBackward(torch.autograd.Function):
    """identity forward, all-reduce backward (column-parallel input)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, dy):
        dy = dy.contiguous()
        dy, _ = _all_reduce(dy, ctx.group)
        return dy, None


def reduce_backward(x, group):
    return _ReduceBackward.apply(x, group)


class _GatherForwardSplitBackward(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, dim, group, grad_scale):
        ctx.dim, ctx.group, ctx.grad_scale = dim, group, grad_scale
        return _gather(x, dim, group)

    @staticmethod
    def backward(ctx, dy):
        dy = _split(dy, ctx.dim, ctx.group)
        if ctx.grad_scale == "up":
            dy = dy * dist.get_world_size(ctx.group)
        elif ctx.grad_scale == "down":
            dy = dy / dist.get_world_size(ctx.group)
        return dy, None, None, None


def gather_forward_split_backward(x, dim, group, grad_scale=None):
    return _GatherForwardSplitBackward.apply(x, dim, group, grad_scale)


class _SplitForwardGatherBackward(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, dim, group, grad_scale):
        ctx.dim, ctx.group, ctx.grad_scale = dim, group, grad_scale
        return _split(x, dim, group)

    @staticmethod
    def backward(ctx, dy):
        dy = _gather(dy.contiguous(), ctx.dim, ctx.group)
        if ctx.grad_scale == "up":
            dy = dy * dist.get_world_size(ctx.group)
        elif ctx.grad_scale == "down":
            dy = dy / dist.get_world_size(ctx.group)
        return dy, None, None, None


def split_forward_gather_backward(x, dim, group, grad_scale=None):
    return _SplitForwardGatherBackward.apply(x, dim, group, grad_scale)


class _GatherForwardReduceScatterBackward(torch.autograd.Function):
    """SP split_gather mode: all-gather sequence in forward, reduce-scatter
    gradient in backward."""

    @staticmethod
    def forward(ctx, x, dim, group):
        ctx.dim, ctx.group = dim, group
        return _gather(x, dim, group)

    @staticmethod
    def backward(ctx, dy):
        return _reduce_scatter(dy.contiguous(), ctx.dim, ctx.group), None, None


def gather_forward_reduce_scatter_backward(x, dim, group):
    return _GatherForwardReduceScatterBackward.apply(x, dim, group)


class _ReduceScatterForwardGatherBackward(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, dim, group):
        ctx.dim, ctx.group = dim, group
        return _reduce_scatter(x.contiguous(), dim, group)

    @staticmethod
    def backward(ctx, dy):
        return _gather(dy.contiguous(), ctx.dim, ctx.group), None, None


def reduce_scatter_forward_gather_backward(x, dim, group):
    return _ReduceScatterForwardGatherBackward.apply(x, dim, group)


class _AllToAll(torch.autograd.Function):
    """Ulysses SP: scatter one dim, gather another; backward is the inverse."""

    @staticmethod
    def forward(ctx, x, group, scatter_dim, gather_dim):
        ctx.group, ctx.scatter_dim, ctx.gather_dim = group, scatter_dim, gather_dim
        return _all_to_all(x, scatter_dim, gather_dim, group)

    @staticmethod
    def backward(ctx, dy):
        dy = _all_to_all(dy.contiguous(), ctx.gather_dim, ctx.scatter_dim, ctx.group)
        return dy, None, None, None


def all_to_all_comm(x, group, scatter_dim=2, gather_dim=1):
    return _AllToAll.apply(x, group, scatter_dim, gather_dim)


# ------------------------------------------------- SP "ring" mode linears
# Ring-pipelined equivalents of the split_gather collectives: instead of a
# monolithic all-gather / reduce-scatter around the matmul, sequence shards
# travel the sp ring one hop per step and each hop's xGMI transfer rides
# under the partial GEMM (reference: _operation.py:418 _ring_as_gather,
# :646 _ring_as_reducescatter). Layout-identical to split_gather; only the
# comm schedule differs.


class _RingGatherLinearCol(torch.autograd.Function):
    """Column-parallel linear with ring-gathered sequence input.

    forward:  y[:, full seq] = allgather_seq(x) @ W^T (+ b), computed one
              shard per ring step.
    backward: dx = ring-reduce-scatter of dY @ W; dW accumulated as x shards
              travel a second ring.
    """

    @staticmethod
    def forward(ctx, x, weight, bias, group, dim):
        import torch.nn.functional as F

        world = dist.get_world_size(group) if group is not None else 1
        if world == 1:
            ctx.save_for_backward(x, weight)
            ctx.meta = (bias is not None, group, dim, 1)
            return F.linear(x, weight, bias)
        from .ring_attn import RingComm

        comm = RingComm(group)
        rank = comm.rank
        shard = x.shape[dim]
        out_shape = list(x.shape)
        out_shape[dim] = shard * world
        out_shape[-1] = weight.shape[0]
        y = x.new_empty(out_shape)
        cur = x.contiguous()
        for step in range(world):
            src = (rank - step) % world
            if step + 1 < world:
                nxt, works = comm.send_recv_async([cur])
            y.narrow(dim, src * shard, shard).copy_(F.linear(cur, weight, bias))
            if step + 1 < world:
                comm.wait(works)
                cur = nxt[0]
        ctx.save_for_backward(x, weight)
        ctx.meta = (bias is not None, group, dim, world)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        use_bias, group, dim, world = ctx.meta
        dy = dy.contiguous()
        if world == 1:
            dy2 = dy.reshape(-1, dy.shape[-1])
            x2 = x.reshape(-1, x.shape[-1])
            db = dy2.sum(0) if use_bias else None
            return dy @ weight, dy2.t() @ x2, db, None, None
        from .ring_attn import RingComm

        comm = RingComm(group)
        rank = comm.rank
        shard = x.shape[dim]

        def contrib(j):
            return (dy.narrow(dim, j * shard, shard) @ weight).contiguous()

        # dx: ring reduce-scatter — the partial that ends on rank r is r's
        # own seq shard; each hop overlaps the next partial GEMM
        b = contrib((rank - 1) % world)
        for t in range(1, world):
            recvs, works = comm.send_recv_async([b])
            c = contrib((rank - t - 1) % world)
            comm.wait(works)
            b = recvs[0] + c
        dx = b

        # dW (+ db): x shards ring-gathered, one partial accumulate per hop
        cur = x.contiguous()
        dw = None
        for step in range(world):
            src = (rank - step) % world
            if step + 1 < world:
                nxt, works = comm.send_recv_async([cur])
            dys = dy.narrow(dim, src * shard, shard).reshape(-1, dy.shape[-1])
            piece = dys.t() @ cur.reshape(-1, x.shape[-1])
            dw = piece if dw is None else dw + piece
            if step + 1 < world:
                comm.wait(works)
                cur = nxt[0]
        db = dy.reshape(-1, dy.shape[-1]).sum(0) if use_bias else None
        return dx, dw, db, None, None


def ring_gather_linear_col(x, weight, bias, group, dim=1):
    return _RingGatherLinearCol.apply(x, weight, bias, group, dim)


class _RingReduceScatterLinearRow(torch.autograd.Function):
    """Row-parallel linear with ring-reduce-scattered sequence output.

    forward:  y_shard = reducescatter_seq(x @ W^T); one partial GEMM per
              ring step, hop under the next GEMM. bias added once at the end.
    backward: dX[:, full seq] = ring-gather(dy) @ W per shard; dW accumulated
              on the same ring pass.
    """

    @staticmethod
    def forward(ctx, x, weight, bias, group, dim):
        import torch.nn.functional as F

        world = dist.get_world_size(group) if group is not None else 1
        ctx.save_for_backward(x, weight)
        if world == 1:
            ctx.meta = (bias is not None, group, dim, 1)
            y = F.linear(x, weight)
            return y + bias if bias is not None else y
        from .ring_attn import RingComm

        comm = RingComm(group)
        rank = comm.rank
        assert x.shape[dim] % world == 0
        shard = x.shape[dim] // world
        ctx.meta = (bias is not None, group, dim, world)

        def contrib(j):
            return F.linear(x.narrow(dim, j * shard, shard), weight).contiguous()

        b = contrib((rank - 1) % world)
        for t in range(1, world):
            recvs, works = comm.send_recv_async([b])
            c = contrib((rank - t - 1) % world)
            comm.wait(works)
            b = recvs[0] + c
        if bias is not None:
            b = b + bias
        return b

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        use_bias, group, dim, world = ctx.meta
        dy = dy.contiguous()
        if world == 1:
            dy2 = dy.reshape(-1, dy.shape[-1])
            x2 = x.reshape(-1, x.shape[-1])
            db = dy2.sum(0) if use_bias else None
            return dy @ weight, dy2.t() @ x2, db, None, None
        from .ring_attn import RingComm

        comm = RingComm(group)
        rank = comm.rank
        shard = dy.shape[dim]
        dx_shape = list(x.shape)
        dx = x.new_empty(dx_shape)
        dw = None
        cur = dy
        # ring-gather dy; per hop: dx slice GEMM + dW partial accumulate
        for step in range(world):
            src = (rank - step) % world
            if step + 1 < world:
                nxt, works = comm.send_recv_async([cur])
            dx.narrow(dim, src * shard, shard).copy_(cur @ weight)
            xs = x.narrow(dim, src * shard, shard).reshape(-1, x.shape[-1])
            piece = cur.reshape(-1, dy.shape[-1]).t() @ xs
            dw = piece if dw is None else dw + piece
            if step + 1 < world:
                comm.wait(works)
                cur = nxt[0]
        # bias is applied once per FULL output: grad sums over every shard,
        # which equals the sum over this rank's shard all-reduced by the
        # ring-gather above (each rank saw all dy shards)
        db = None
        if use_bias:
            db = dy.reshape(-1, dy.shape[-1]).sum(0)
            dist.all_reduce(db, group=group)
        return dx, dw, db, None, None


def ring_reducescatter_linear_row(x, weight, bias, group, dim=1):
    return _RingReduceScatterLinearRow.apply(x, weight, bias, group, dim)
