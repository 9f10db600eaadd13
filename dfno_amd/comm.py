"""Distributed tensor primitives over RCCL/xGMI (torch.distributed).

MI355X-native replacements for the DistDL nn primitives the reference uses
(``dnn.Broadcast`` / ``dnn.SumReduce`` / ``dnn.Repartition`` /
``dnn.DistributedTranspose``, see /root/reference/dfno/dfno.py:41-42,99-102 and
/root/reference/dfno/loss.py:17-18).  Each primitive is a
``torch.autograd.Function`` whose backward is the adjoint collective
(SURVEY.md K8-K10, K13):

* ``Broadcast``   : root -> all     (adjoint: sum-reduce to root)
* ``SumReduce``   : all  -> root    (adjoint: broadcast from root)
* ``Repartition`` : block re-decomposition between two Cartesian partitions,
  executed as a precomputed block-intersection plan of grouped
  ``isend``/``irecv`` (RCCL grouped ncclSend/ncclRecv over the 7 xGMI
  point-to-point links — an all-to-all-v, not a ring).

Plans are deduplicated in a registry keyed by (P_src, P_dst, global shape),
fixing the per-block plan proliferation of the reference.
Complex tensors travel as their real views (RCCL has no complex dtype).
Zero-volume tensors (inactive or empty blocks) are handled throughout.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np
import torch
import torch.distributed as dist

from .partition import (
    Partition,
    block_bounds,
    is_distributed,
    world_rank,
    zero_volume_tensor,
)

__all__ = [
    "Broadcast",
    "SumReduce",
    "Repartition",
    "DistributedTranspose",
    "AllReduceSum",
    "ZeroVolumeCorrectorFunction",
    "all_reduce_",
    "repartition_issue",
    "repartition_complete",
    "begin_chain",
    "end_chain",
    "reset_chain",
]


# ---------------------------------------------------------------------------
# helpers
# ---------------------------------------------------------------------------

def _as_real(x: torch.Tensor) -> torch.Tensor:
    """View a (contiguous) tensor as real for communication."""
    if x.is_complex():
        return torch.view_as_real(x)
    return x


def _comm_clone(x: torch.Tensor) -> torch.Tensor:
    """Contiguous clone suitable for p2p communication."""
    return _as_real(x.contiguous()).contiguous()


def all_reduce_(t: torch.Tensor, P: Partition, op: str = "sum") -> torch.Tensor:
    """In-place allreduce over a partition (no autograd). Utility for
    normalisation / metrics (reference K11)."""
    if is_distributed() and P.active and P.size > 1:
        red = {"sum": dist.ReduceOp.SUM, "min": dist.ReduceOp.MIN,
               "max": dist.ReduceOp.MAX}[op]
        if t.is_complex():
            r = torch.view_as_real(t)
            dist.all_reduce(r, op=red, group=P.group)
        else:
            dist.all_reduce(t, op=red, group=P.group)
    return t


# ---------------------------------------------------------------------------
# comm ordering chain
# ---------------------------------------------------------------------------
#
# Matched collectives must leave the autograd engine in the SAME order on
# every rank.  With a purely sequential graph that holds by construction,
# but parallel branches (the channel-chunked pencil pipeline, the loss's two
# SumReduces) let the engine interleave comm backward nodes differently on
# ranks whose local graphs differ (zero-volume ranks have fewer compute
# nodes) — observed deadlock: two ranks parked in a Broadcast-adjoint reduce
# while two waited on a repartition-adjoint p2p.  Fix: every comm Function
# takes and emits a zero-size ordering token; the token chain adds a graph
# edge between consecutive collectives, so backward visits them in exactly
# reverse forward order on every rank.  The chain is reset at each model
# forward (nn/fno.py) and spans model + loss.

_CHAIN = {"tok": None, "active": False}


def chain_token(ref: torch.Tensor) -> torch.Tensor:
    """Chain link for a comm op on ``ref``.

    The chain is SCOPED to one model forward (begin_chain/end_chain in
    DistributedFNONd.forward): only there do parallel graph branches (the
    channel-chunked pipeline) exist, and only there may tokens link comm
    ops — an unscoped chain would thread tokens across independently
    backwarded graphs (two models, collector gathers), making the second
    backward traverse an already-freed first graph.  Within the scope,
    only ops whose backward will RUN (grad mode + ref requires grad) join;
    ``ref.requires_grad`` is uniform across ranks at every model comm site,
    which the ordering argument requires.
    """
    if not (_CHAIN["active"] and torch.is_grad_enabled() and ref.requires_grad):
        return torch.zeros(0, device=ref.device)
    t = _CHAIN["tok"]
    if t is None or t.device != ref.device or not t.requires_grad:
        t = torch.zeros(0, device=ref.device, requires_grad=True)
    return t


def set_chain(t: torch.Tensor) -> None:
    # advance the chain only when the new link carries grad history;
    # non-grad comm must not sever the ordering between grad comm ops
    if _CHAIN["active"] and t is not None and t.grad_fn is not None:
        _CHAIN["tok"] = t


def begin_chain() -> None:
    _CHAIN["tok"] = None
    _CHAIN["active"] = True


def end_chain() -> None:
    _CHAIN["tok"] = None
    _CHAIN["active"] = False


def reset_chain() -> None:
    end_chain()


# ---------------------------------------------------------------------------
# Broadcast (root-stored weights -> all ranks each forward; adjoint = reduce)
# ---------------------------------------------------------------------------

class _BroadcastFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, tok: torch.Tensor, module: "Broadcast"):
        ctx.module = module
        P = module.P_dst
        if not (is_distributed() and P.active and P.size > 1):
            # serial / single-member partition: identity
            ctx.was_root = True
            return x.clone(), tok.new_empty(0)
        root_world = module.P_src.ranks[0]
        is_root = world_rank() == root_world
        ctx.was_root = is_root

        meta = module._meta
        if meta is None:
            # one-time metadata exchange: shape + dtype of the payload
            obj = [None]
            if is_root:
                obj = [(tuple(x.shape), x.dtype)]
            dist.broadcast_object_list(obj, src=root_world, group=P.group)
            meta = obj[0]
            module._meta = meta
        shape, dtype = meta
        if is_root and (tuple(x.shape) != shape or x.dtype != dtype):
            raise RuntimeError(
                f"Broadcast: payload changed from {shape}/{dtype} to "
                f"{tuple(x.shape)}/{x.dtype}; a Broadcast module is bound to "
                "one payload signature (non-root ranks cache it) - build a "
                "new module for a different tensor")
        if is_root:
            buf = _comm_clone(x)
        else:
            ref = torch.empty(shape, dtype=dtype, device=module._device_of(x))
            buf = _as_real(ref).contiguous()
        dist.broadcast(buf, src=root_world, group=P.group)
        if dtype.is_complex:
            out = torch.view_as_complex(buf)
        else:
            out = buf
        return out, tok.new_empty(0)

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor, grad_tok):
        module: Broadcast = ctx.module
        P = module.P_dst
        tg = torch.empty(0, device=grad_out.device)
        if not (is_distributed() and P.active and P.size > 1):
            return grad_out, tg, None
        root_world = module.P_src.ranks[0]
        buf = _comm_clone(grad_out)
        dist.reduce(buf, dst=root_world, op=dist.ReduceOp.SUM, group=P.group)
        if ctx.was_root:
            if grad_out.is_complex():
                buf = torch.view_as_complex(buf)
            return buf, tg, None
        # non-root: parameter is a zero-volume placeholder
        return (zero_volume_tensor(device=grad_out.device, dtype=grad_out.dtype),
                tg, None)


class Broadcast(torch.nn.Module):
    """Broadcast a root-stored tensor to every rank of ``P_dst``.

    Reference counterpart: ``dnn.Broadcast(P_root, P_x)``
    (/root/reference/dfno/dfno.py:41-42).
    """

    def __init__(self, P_src: Partition, P_dst: Partition):
        super().__init__()
        self.P_src = P_src
        self.P_dst = P_dst
        self._meta: Optional[Tuple[Tuple[int, ...], torch.dtype]] = None
        self._device = None

    def _device_of(self, x: torch.Tensor):
        if self._device is None:
            if torch.cuda.is_available():
                self._device = torch.device("cuda", torch.cuda.current_device())
            else:
                self._device = torch.device("cpu")
        # non-root input is zero-volume but carries the right device
        if x.numel() == 0 and x.device is not None:
            return x.device
        return x.device

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        out, tok = _BroadcastFn.apply(x, chain_token(x), self)
        set_chain(tok)
        return out


# ---------------------------------------------------------------------------
# SumReduce (all -> root; adjoint = broadcast)
# ---------------------------------------------------------------------------

class _SumReduceFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, tok: torch.Tensor, module: "SumReduce"):
        ctx.module = module
        ctx.in_shape = tuple(x.shape)
        ctx.in_dtype = x.dtype
        ctx.in_device = x.device
        P = module.P_src
        if not (is_distributed() and P.active and P.size > 1):
            return x.clone(), tok.new_empty(0)
        root_world = module.P_dst.ranks[0]
        buf = _comm_clone(x)
        dist.reduce(buf, dst=root_world, op=dist.ReduceOp.SUM, group=P.group)
        if world_rank() == root_world:
            if x.is_complex():
                buf = torch.view_as_complex(buf)
            return buf, tok.new_empty(0)
        return zero_volume_tensor(device=x.device, dtype=x.dtype), tok.new_empty(0)

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor, grad_tok):
        module: SumReduce = ctx.module
        P = module.P_src
        tg = torch.empty(0, device=ctx.in_device)
        if not (is_distributed() and P.active and P.size > 1):
            return grad_out, tg, None
        root_world = module.P_dst.ranks[0]
        if world_rank() == root_world:
            buf = _comm_clone(grad_out)
        else:
            ref = torch.empty(ctx.in_shape, dtype=ctx.in_dtype, device=ctx.in_device)
            buf = _as_real(ref).contiguous()
        dist.broadcast(buf, src=root_world, group=P.group)
        if ctx.in_dtype.is_complex:
            buf = torch.view_as_complex(buf)
        return buf, tg, None


class SumReduce(torch.nn.Module):
    """Sum-reduce over ``P_src`` onto the root of ``P_dst``.

    Reference counterpart: ``dnn.SumReduce(P_x, P_0)``
    (/root/reference/dfno/loss.py:17-18).
    """

    def __init__(self, P_src: Partition, P_dst: Partition):
        super().__init__()
        self.P_src = P_src
        self.P_dst = P_dst

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        out, tok = _SumReduceFn.apply(x, chain_token(x), self)
        set_chain(tok)
        return out


# ---------------------------------------------------------------------------
# AllReduce (sum) with autograd (adjoint of allreduce-sum is allreduce-sum)
# ---------------------------------------------------------------------------

class _AllReduceSumFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, tok: torch.Tensor, P: Partition):
        ctx.P = P
        if not (is_distributed() and P.active and P.size > 1):
            return x.clone(), tok.new_empty(0)
        buf = _comm_clone(x)
        dist.all_reduce(buf, op=dist.ReduceOp.SUM, group=P.group)
        if x.is_complex():
            buf = torch.view_as_complex(buf)
        return buf, tok.new_empty(0)

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor, grad_tok):
        P = ctx.P
        tg = torch.empty(0, device=grad_out.device)
        if not (is_distributed() and P.active and P.size > 1):
            return grad_out, tg, None
        buf = _comm_clone(grad_out)
        dist.all_reduce(buf, op=dist.ReduceOp.SUM, group=P.group)
        if grad_out.is_complex():
            buf = torch.view_as_complex(buf)
        return buf, tg, None


class AllReduceSum(torch.nn.Module):
    """Autograd-aware sum allreduce over a partition."""

    def __init__(self, P: Partition):
        super().__init__()
        self.P = P

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        out, tok = _AllReduceSumFn.apply(x, chain_token(x), self.P)
        set_chain(tok)
        return out


# ---------------------------------------------------------------------------
# Repartition (block-redistribution; the pencil-FFT all-to-all)
# ---------------------------------------------------------------------------

class _RepartitionPlan:
    """Precomputed block-intersection send/recv schedule for one
    (P_src, P_dst, global_shape) triple.

    send[i] = (peer_world_rank, src_local_slices)
    recv[i] = (peer_world_rank, dst_local_slices, piece_shape)
    ``local_copy``: the overlapping box this rank keeps (src slices, dst
    slices) when it is active in both partitions.
    """

    def __init__(self, P_src: Partition, P_dst: Partition, gshape: Tuple[int, ...]):
        self.P_src = P_src
        self.P_dst = P_dst
        self.gshape = gshape
        me = world_rank()
        dim = len(gshape)

        self.out_shape: Optional[List[int]] = None
        self.sends: List[Tuple[int, Tuple[slice, ...]]] = []
        self.recvs: List[Tuple[int, Tuple[slice, ...], Tuple[int, ...]]] = []
        self.local_copy: Optional[Tuple[Tuple[slice, ...], Tuple[slice, ...]]] = None

        src_bounds = None
        if P_src.active:
            src_bounds = block_bounds(P_src, gshape, P_src.rank)
        dst_bounds = None
        if P_dst.active:
            dst_bounds = block_bounds(P_dst, gshape, P_dst.rank)
            self.out_shape = [b - a for a, b in dst_bounds]

        def intersect(a, b):
            box = []
            for (a0, a1), (b0, b1) in zip(a, b):
                lo, hi = max(a0, b0), min(a1, b1)
                if hi <= lo:
                    return None
                box.append((lo, hi))
            return box

        if P_src.active:
            # what this rank sends to every dst block
            for q in range(P_dst.size):
                qb = block_bounds(P_dst, gshape, q)
                box = intersect(src_bounds, qb)
                if box is None:
                    continue
                peer = P_dst.ranks[q]
                sl = tuple(slice(lo - s0, hi - s0) for (lo, hi), (s0, _) in zip(box, src_bounds))
                if peer == me:
                    dsl = tuple(slice(lo - d0, hi - d0) for (lo, hi), (d0, _) in zip(box, dst_bounds))
                    self.local_copy = (sl, dsl)
                else:
                    self.sends.append((peer, sl))
        if P_dst.active:
            for q in range(P_src.size):
                qb = block_bounds(P_src, gshape, q)
                box = intersect(dst_bounds, qb)
                if box is None:
                    continue
                peer = P_src.ranks[q]
                if peer == me:
                    continue  # handled by local_copy
                dsl = tuple(slice(lo - d0, hi - d0) for (lo, hi), (d0, _) in zip(box, dst_bounds))
                shp = tuple(hi - lo for lo, hi in box)
                self.recvs.append((peer, dsl, shp))
        # Deterministic peer order (matched on both sides by construction).
        self.sends.sort(key=lambda t: t[0])
        self.recvs.sort(key=lambda t: t[0])

        # per-(dtype, device) packed-exchange descriptors (built lazily)
        self._packed: Dict[Tuple, "_PackedMeta"] = {}

        # Identity detection: this rank keeps its whole block and exchanges
        # nothing -> the repartition is a no-op here (e.g. P_x == P_m for the
        # two-phase partition (1,1,1,N,1,1)).  Skipping it avoids a full
        # tensor copy per call on the hot path.
        self.is_identity = False
        if not self.sends and not self.recvs:
            if self.out_shape is None:
                self.is_identity = not P_src.active  # zero-volume in, out
            elif self.local_copy is not None:
                ssl, dsl = self.local_copy
                full = all(
                    s.start == 0 and s.stop == e and d.start == 0 and d.stop == e
                    for s, d, e in zip(ssl, dsl, self.out_shape))
                src_shape = [b - a for a, b in src_bounds] if src_bounds else None
                self.is_identity = full and src_shape == self.out_shape


_PLAN_REGISTRY: Dict[Tuple, _RepartitionPlan] = {}


def _get_plan(P_src: Partition, P_dst: Partition, gshape: Tuple[int, ...]) -> _RepartitionPlan:
    key = (P_src.ranks, tuple(P_src.shape), P_dst.ranks, tuple(P_dst.shape), gshape)
    plan = _PLAN_REGISTRY.get(key)
    if plan is None:
        plan = _RepartitionPlan(P_src, P_dst, gshape)
        _PLAN_REGISTRY[key] = plan
    return plan


_PACK_REC = 20   # longs per piece record (csrc/pack.hip)


def _box_record(shape: Tuple[int, ...], box, wpe: int, flat_off: int):
    """One pack descriptor record for a box of a contiguous tensor.

    ``shape``: element-space local tensor shape; ``box``: per-dim (start,
    stop); ``wpe``: words per element (2 for complex viewed as real words).
    Contiguous inner runs are merged host-side so the kernel's divmod chain
    is 2-3 deep in practice.
    """
    nd = len(shape)
    estr = [1] * nd
    for d in range(nd - 2, -1, -1):
        estr[d] = estr[d + 1] * shape[d + 1]
    lens = [hi - lo for lo, hi in box]
    numel = 1
    for l in lens:
        numel *= l
    numel *= wpe
    dims = [int(l) for l in lens] + ([wpe] if wpe > 1 else [])
    strs = [estr[d] * wpe for d in range(nd)] + ([1] if wpe > 1 else [])
    off = sum(lo * estr[d] for d, (lo, _) in enumerate(box)) * wpe
    # merge contiguous runs, inner -> outer
    blocks = []  # (len, stride), innermost first
    for l, s in zip(reversed(dims), reversed(strs)):
        if l == 1:
            continue
        if blocks and s == blocks[-1][0] * blocks[-1][1]:
            blocks[-1] = (blocks[-1][0] * l, blocks[-1][1])
        else:
            blocks.append((l, s))
    if not blocks:
        blocks = [(1, 1)]
    blocks.reverse()  # outer -> inner
    assert len(blocks) <= 8, "pack box rank > 8 after merging"
    mdims = [l for l, _ in blocks] + [1] * (8 - len(blocks))
    mstrs = [s for _, s in blocks] + [0] * (8 - len(blocks))
    return [flat_off, off, numel, len(blocks), *mdims, *mstrs], numel


class _PackedMeta:
    """Cached device descriptors for one (plan, dtype, device)."""

    __slots__ = ("send_desc", "send_total", "send_ranges", "send_max",
                 "recv_desc", "recv_total", "recv_ranges", "recv_max",
                 "word_dtype")

    def __init__(self, plan: "_RepartitionPlan", dtype: torch.dtype, device):
        wpe = 2 if dtype.is_complex else 1
        if dtype in (torch.complex128, torch.float64):
            self.word_dtype = torch.float64
        elif dtype == torch.bfloat16:
            self.word_dtype = torch.bfloat16   # 2-byte words (pack.hip ushort)
        else:
            self.word_dtype = torch.float32
        src_shape = None
        if plan.P_src.active:
            sb = block_bounds(plan.P_src, plan.gshape, plan.P_src.rank)
            src_shape = tuple(b - a for a, b in sb)

        recs, ranges, off, mx = [], [], 0, 1
        for peer, ssl in plan.sends:
            box = [(s.start or 0, s.stop) for s in ssl]
            rec, n = _box_record(src_shape, box, wpe, off)
            recs.append(rec)
            ranges.append((peer, off, n))
            off += n
            mx = max(mx, n)
        self.send_desc = (torch.tensor(recs, dtype=torch.int64, device=device)
                          if recs else None)
        self.send_total, self.send_ranges, self.send_max = off, ranges, mx

        recs, ranges, off, mx = [], [], 0, 1
        for peer, dsl, shp in plan.recvs:
            box = [(s.start or 0, s.stop) for s in dsl]
            rec, n = _box_record(tuple(plan.out_shape), box, wpe, off)
            recs.append(rec)
            ranges.append((peer, off, n))
            off += n
            mx = max(mx, n)
        self.recv_desc = (torch.tensor(recs, dtype=torch.int64, device=device)
                          if recs else None)
        self.recv_total, self.recv_ranges, self.recv_max = off, ranges, mx


def _issue_plan_packed(plan: "_RepartitionPlan", x: torch.Tensor,
                       out: torch.Tensor, dtype: torch.dtype, device):
    """GPU packed exchange: one pack kernel -> grouped isend/irecv on slices
    of a flat staging buffer -> one unpack kernel at completion.  Returns
    None when the extension is unavailable (caller falls back to slicing)."""
    from . import _ext
    ext = _ext.get(required=False)
    if ext is None:
        return None
    if dtype not in (torch.float32, torch.float64, torch.bfloat16,
                     torch.complex64, torch.complex128):
        return None   # unsupported word size: slicing path
    if not x.is_contiguous():
        x = x.contiguous()

    key = (dtype, x.device.index)
    meta = plan._packed.get(key)
    if meta is None:
        meta = _PackedMeta(plan, dtype, device)
        plan._packed[key] = meta

    word = meta.word_dtype
    # reinterpret the contiguous storage as flat words (complex -> real pairs)
    x_words = _as_real(x).reshape(-1) if x.numel() else torch.empty(0, dtype=word, device=device)
    out_words = _as_real(out).reshape(-1) if out.numel() else torch.empty(0, dtype=word, device=device)

    ops = []
    flat_send = None
    if meta.send_desc is not None:
        flat_send = torch.empty(meta.send_total, dtype=word, device=device)
        ext.pack_boxes(x_words, flat_send, meta.send_desc, meta.send_max)
        for peer, off, n in meta.send_ranges:
            ops.append(dist.P2POp(dist.isend, flat_send.narrow(0, off, n), peer))
    flat_recv = None
    if meta.recv_desc is not None:
        flat_recv = torch.empty(meta.recv_total, dtype=word, device=device)
        for peer, off, n in meta.recv_ranges:
            ops.append(dist.P2POp(dist.irecv, flat_recv.narrow(0, off, n), peer))

    if plan.local_copy is not None:
        ssl, dsl = plan.local_copy
        out[dsl] = x[ssl]

    reqs = dist.batch_isend_irecv(ops) if ops else ()

    unpack = None
    if flat_recv is not None:
        def unpack(fr=flat_recv, ow=out_words, d=meta.recv_desc, m=meta.recv_max):
            ext.unpack_boxes(fr, ow, d, m)
    return _PendingRepart(plan, out, reqs, (), dtype,
                          send_bufs=(flat_send,) if flat_send is not None else (),
                          unpack=unpack)


class _PendingRepart:
    """In-flight repartition: sends/recvs issued, completion deferred.

    The issue/complete split is the comm/compute overlap hook (SURVEY.md K9,
    VERDICT.md round-1 item 1): issue ALL chunk exchanges first, then
    complete+compute chunk by chunk — on RCCL the exchanges run on the NCCL
    stream while the default stream computes earlier chunks; on gloo the
    transfers progress on background threads during host compute.
    """

    __slots__ = ("plan", "out", "reqs", "recv_bufs", "dtype", "done",
                 "send_bufs", "_unpack")

    def __init__(self, plan, out, reqs, recv_bufs, dtype,
                 send_bufs=(), unpack=None):
        self.plan = plan
        self.out = out
        self.reqs = reqs
        self.recv_bufs = recv_bufs
        self.dtype = dtype
        self.done = False
        self.send_bufs = send_bufs
        self._unpack = unpack

    def complete(self) -> torch.Tensor:
        if self.done:
            return self.out
        for r in self.reqs:
            r.wait()
        if self._unpack is not None:
            self._unpack()
        else:
            for buf, dsl, shp in self.recv_bufs:
                if self.dtype.is_complex:
                    self.out[dsl] = torch.view_as_complex(buf)
                else:
                    self.out[dsl] = buf
        self.done = True
        # release send/recv staging
        self.reqs = ()
        self.recv_bufs = ()
        self.send_bufs = ()
        self._unpack = None
        return self.out


def _issue_plan(plan: _RepartitionPlan, x: torch.Tensor,
                dtype: torch.dtype, device) -> _PendingRepart:
    """Pack + post all sends/recvs of a repartition plan; defer completion."""
    P_src, P_dst = plan.P_src, plan.P_dst

    if plan.out_shape is None:
        out = zero_volume_tensor(device=device, dtype=dtype)
    else:
        out = torch.zeros(plan.out_shape, dtype=dtype, device=device)

    if not is_distributed() or (P_src.size == 1 and P_dst.size == 1):
        if plan.local_copy is not None:
            ssl, dsl = plan.local_copy
            out[dsl] = x[ssl]
        return _PendingRepart(plan, out, (), (), dtype)

    if x.is_cuda and (plan.sends or plan.recvs or plan.local_copy):
        pending = _issue_plan_packed(plan, x, out, dtype, device)
        if pending is not None:
            return pending

    ops = []
    send_bufs = []
    recv_bufs = []
    for peer, ssl in plan.sends:
        buf = _comm_clone(x[ssl])
        send_bufs.append(buf)
        ops.append(dist.P2POp(dist.isend, buf, peer))
    for peer, dsl, shp in plan.recvs:
        ref = torch.empty(shp, dtype=dtype, device=device)
        buf = _as_real(ref).contiguous()
        recv_bufs.append((buf, dsl, shp))
        ops.append(dist.P2POp(dist.irecv, buf, peer))

    if plan.local_copy is not None:
        ssl, dsl = plan.local_copy
        out[dsl] = x[ssl]

    reqs = dist.batch_isend_irecv(ops) if ops else ()
    # send buffers stay alive until completion (explicitly referenced)
    return _PendingRepart(plan, out, reqs, recv_bufs, dtype, send_bufs=send_bufs)


def _execute_plan(plan: _RepartitionPlan, x: torch.Tensor,
                  dtype: torch.dtype, device) -> torch.Tensor:
    """Run a repartition plan on tensor ``x`` (zero-volume on inactive src)."""
    return _issue_plan(plan, x, dtype, device).complete()


# ---------------------------------------------------------------------------
# split-phase (issue/complete) repartition for comm/compute overlap
# ---------------------------------------------------------------------------
#
# The pipelined pencil chain (nn/block.py) issues ALL channel-chunk exchanges
# before completing/computing any of them: on RCCL the exchanges proceed on
# the NCCL stream while the default stream transforms earlier chunks; on gloo
# they progress on background threads under host compute.  The split is two
# autograd Functions linked by a zero-volume token; the in-flight state lives
# in a ticket registry.
#
# BACKWARD runs the adjoint exchange SYNCHRONOUSLY inside the complete-fn's
# backward node (the finished gradient is handed to the issue-fn's backward
# via the registry).  Splitting the adjoint's post/wait across two backward
# nodes deadlocks: the autograd engine schedules per-rank-different compute
# nodes (zero-volume ranks have fewer) between them, so collectives leave
# graph order differently on different ranks (observed: two ranks parked in
# a Broadcast-adjoint reduce while two waited on the adjoint p2p).  Forward
# overlap — the measured win — is unaffected: its issue/complete interleave
# is explicit, identical code on every rank.

import itertools as _itertools

_TICKETS: Dict[int, "_PendingRepart"] = {}
_ADJ_GRADS: Dict[int, torch.Tensor] = {}
_ticket_counter = _itertools.count(1)


class _RepartIssueFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, tok: torch.Tensor, module: "Repartition",
                gshape: Tuple[int, ...], ticket: int) -> torch.Tensor:
        plan = _get_plan(module.P_src, module.P_dst, gshape)
        _TICKETS[ticket] = _issue_plan(plan, x, x.dtype, x.device)
        ctx.ticket = ticket
        return torch.empty(0, dtype=x.dtype, device=x.device)

    @staticmethod
    def backward(ctx, g_token: torch.Tensor):
        gx = _ADJ_GRADS.pop(ctx.ticket)
        return gx, torch.empty(0, device=gx.device), None, None, None


class _RepartCompleteFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, token: torch.Tensor, tok: torch.Tensor,
                module: "Repartition", gshape: Tuple[int, ...],
                ticket: int):
        ctx.module = module
        ctx.gshape = gshape
        ctx.ticket = ticket
        return _TICKETS.pop(ticket).complete(), tok.new_empty(0)

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor, grad_tok):
        module: Repartition = ctx.module
        plan = _get_plan(module.P_dst, module.P_src, ctx.gshape)
        _ADJ_GRADS[ctx.ticket] = _issue_plan(
            plan, grad_out.contiguous(), grad_out.dtype,
            grad_out.device).complete()
        token_grad = torch.empty(0, dtype=grad_out.dtype, device=grad_out.device)
        return token_grad, torch.empty(0, device=grad_out.device), None, None, None


def repartition_issue(R: "Repartition", x: torch.Tensor, global_shape):
    """Post the exchange for ``x``; returns a handle for repartition_complete."""
    g = tuple(int(s) for s in global_shape)
    plan = _get_plan(R.P_src, R.P_dst, g)
    if plan.is_identity:
        return ("id", x)
    t = next(_ticket_counter)
    token = _RepartIssueFn.apply(x, chain_token(x), R, g, t)
    set_chain(token)  # the issue's own output token doubles as the chain link
    return ("tok", token, R, g, t)


def repartition_complete(handle) -> torch.Tensor:
    if handle[0] == "id":
        return handle[1]
    _, token, R, g, t = handle
    out, tok = _RepartCompleteFn.apply(token, chain_token(token), R, g, t)
    set_chain(tok)
    return out


class _RepartitionFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, tok: torch.Tensor, module: "Repartition",
                gshape: Tuple[int, ...]):
        ctx.module = module
        ctx.gshape = gshape
        ctx.dtype = x.dtype
        ctx.device = x.device
        plan = _get_plan(module.P_src, module.P_dst, gshape)
        return _execute_plan(plan, x, x.dtype, x.device), tok.new_empty(0)

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor, grad_tok):
        module: Repartition = ctx.module
        # adjoint = reversed repartition on the gradient
        plan = _get_plan(module.P_dst, module.P_src, ctx.gshape)
        gx = _execute_plan(plan, grad_out.contiguous(), ctx.dtype, ctx.device)
        return gx, torch.empty(0, device=ctx.device), None, None


class Repartition(torch.nn.Module):
    """Re-decompose a block-distributed tensor from ``P_src`` to ``P_dst``.

    Reference counterpart: ``dnn.Repartition`` / ``dnn.DistributedTranspose``
    (/root/reference/dfno/dfno.py:99-102, .../test_two_phase.py:21-23).

    The global shape is taken from (in priority order): the ``global_shape``
    argument to :meth:`forward`, the constructor's ``global_shape``, or a
    one-time gather of local shapes over the union group (cached per local
    input shape).
    """

    def __init__(self, P_src: Partition, P_dst: Partition,
                 global_shape: Optional[Sequence[int]] = None,
                 preserve_batch: bool = False):
        super().__init__()
        self.P_src = P_src
        self.P_dst = P_dst
        self._gshape = tuple(int(s) for s in global_shape) if global_shape is not None else None
        self._inferred: Dict[Tuple[int, ...], Tuple[int, ...]] = {}

    def _infer_gshape(self, x: torch.Tensor) -> Tuple[int, ...]:
        key = tuple(x.shape)
        hit = self._inferred.get(key)
        if hit is not None:
            return hit
        P = self.P_src
        if not is_distributed() or P.size == 1:
            g = key
        else:
            # Gather every src rank's (index, local shape); reconstruct global
            # extents from blocks along each axis at index 0 of other axes.
            obj = (tuple(int(i) for i in P.index), key) if P.active else None
            gathered: List = [None] * dist.get_world_size()
            dist.all_gather_object(gathered, obj)
            infos = [o for o in gathered if o is not None]
            dim = P.dim
            g = []
            for d in range(dim):
                tot = 0
                for idx, shp in infos:
                    if all(idx[e] == 0 for e in range(dim) if e != d):
                        tot += shp[d]
                g.append(tot)
            g = tuple(g)
        self._inferred[key] = g
        return g

    def forward(self, x: torch.Tensor, global_shape: Optional[Sequence[int]] = None) -> torch.Tensor:
        if global_shape is not None:
            g = tuple(int(s) for s in global_shape)
        elif self._gshape is not None:
            g = self._gshape
        else:
            g = self._infer_gshape(x)
        plan = _get_plan(self.P_src, self.P_dst, g)
        if plan.is_identity:
            return x
        out, tok = _RepartitionFn.apply(x, chain_token(x), self, g)
        set_chain(tok)
        return out


# DistDL exposes the same op under a second name; the reference's NS trainer
# uses it (/root/reference/training/navier_stokes/experiment_navier_stokes.py:91-93).
DistributedTranspose = Repartition


# ---------------------------------------------------------------------------
# ZeroVolumeCorrector (reference: distdl.functional.ZeroVolumeCorrectorFunction,
# used at /root/reference/dfno/loss.py:35)
# ---------------------------------------------------------------------------

class ZeroVolumeCorrectorFunction(torch.autograd.Function):
    """Make a distributed scalar loss backward-able on every rank.

    On ranks holding a zero-volume tensor, returns a detached-from-value but
    graph-connected scalar 0 so that ``loss.backward()`` can be called
    SPMD-style everywhere; backward returns a zero-volume gradient.
    """

    @staticmethod
    def forward(ctx, x: torch.Tensor) -> torch.Tensor:
        ctx.zero_volume = x.numel() == 0
        ctx.in_shape = tuple(x.shape)
        ctx.in_dtype = x.dtype
        ctx.in_device = x.device
        if ctx.zero_volume:
            return torch.zeros((), dtype=x.dtype if x.dtype.is_floating_point else torch.float32,
                               device=x.device)
        return x.clone()

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        if ctx.zero_volume:
            return torch.empty(ctx.in_shape, dtype=ctx.in_dtype, device=ctx.in_device)
        return grad_out
