"""PyTorch collective op API.

Reference: horovod/torch/mpi_ops.py (1333 LoC — allreduce/allgather/
broadcast/alltoall/reducescatter families with async/in-place/grouped
variants, autograd functions, sparse allreduce, join/barrier).  This is the
same user surface on the MI355X-native core: one templated native entry point
per op (no per-dtype symbols), wire-dtype compression fused into the pack
kernels, RCCL over xGMI underneath.
"""
from contextlib import contextmanager

import torch

from horovod_amd import _core
from horovod_amd.common.basics import HorovodBasics
from horovod_amd.common.process_sets import (ProcessSet, global_process_set,
                                             add_process_set,  # noqa: F401
                                             remove_process_set)  # noqa: F401

from horovod_amd.common.exceptions import HorovodInternalError

_basics = HorovodBasics()


def _translate_error(fn, *args, **kwargs):
    """Native-core errors tagged HorovodInternalError become the typed
    exception the elastic retry loop catches."""
    try:
        return fn(*args, **kwargs)
    except RuntimeError as e:
        if "HorovodInternalError" in str(e):
            raise HorovodInternalError(str(e)) from None
        raise

init = _basics.init
shutdown = _basics.shutdown
is_initialized = _basics.is_initialized
rank = _basics.rank
size = _basics.size
local_rank = _basics.local_rank
local_size = _basics.local_size
cross_rank = _basics.cross_rank
cross_size = _basics.cross_size
mpi_threads_supported = _basics.mpi_threads_supported
mpi_enabled = _basics.mpi_enabled
mpi_built = _basics.mpi_built
gloo_enabled = _basics.gloo_enabled
gloo_built = _basics.gloo_built
nccl_built = _basics.nccl_built
ddl_built = _basics.ddl_built
ccl_built = _basics.ccl_built
cuda_built = _basics.cuda_built
rocm_built = _basics.rocm_built
is_homogeneous = _basics.is_homogeneous

# ReduceOp codes (must match hvd::ReduceOp in csrc/common.h)
Average = 0
Sum = 1
Adasum = 2
Min = 3
Max = 4
Product = 5

_NULL_NAME_COUNTER = 0

# torch dtype -> wire DataType code (csrc/common.h DataType)
_DTYPE_CODES = {
    torch.uint8: 0, torch.int8: 1, torch.int32: 2, torch.int64: 3,
    torch.float16: 4, torch.float32: 5, torch.float64: 6, torch.bool: 7,
    torch.bfloat16: 8, torch.int16: 10,
}


def _dtype_code(dtype):
    try:
        return _DTYPE_CODES[dtype]
    except KeyError:
        raise ValueError(f"horovod_amd: unsupported dtype {dtype}")


def _dense_ok(t):
    """True when the tensor occupies a dense (non-overlapping) block of
    memory, possibly with permuted strides (e.g. channels_last).  Dense
    tensors are raw-copyable: every rank shares the identical layout, so the
    pack kernels and RCCL operate in memory order consistently."""
    if t.is_contiguous():
        return True
    if t.dim() == 4 and t.is_contiguous(memory_format=torch.channels_last):
        return True
    if t.dim() == 5 and t.is_contiguous(memory_format=torch.channels_last_3d):
        return True
    return False


def _next_name(prefix):
    global _NULL_NAME_COUNTER
    _NULL_NAME_COUNTER += 1
    return f"{prefix}.noname.{_NULL_NAME_COUNTER}"


def _set_id(process_set):
    ps_id = process_set.process_set_id
    if ps_id is None:
        raise ValueError(
            "Attempted to use a ProcessSet that has not been registered via "
            "hvd.add_process_set().")
    return ps_id


def _set_size(process_set):
    if process_set.process_set_id == 0:
        return size()
    return len(process_set.ranks)


def _resolve_scales(op, average, prescale_factor, postscale_factor, process_set):
    """Map (op, average) + user scale factors onto SUM + pre/post scales.

    The reference applies Average via pre/postscale on GPU (mpi_ops_v2.cc
    62-114, DivideInPlace for ROCm); here averaging always folds into the
    unpack kernel's postscale — zero extra passes.
    """
    if average is not None:
        if op is not None:
            raise ValueError('The op parameter supersedes average. Please '
                             'provide only one of them.')
        op = Average if average else Sum
    if op is None:
        op = Average
    true_op = op
    if op == Average:
        true_op = Sum
        postscale_factor = postscale_factor / _set_size(process_set)
    return true_op, op, prescale_factor, postscale_factor


class _HandleInfo:
    __slots__ = ("native", "outputs_n", "kind", "post_divisor")

    def __init__(self, native, outputs_n=1, kind="op", post_divisor=None):
        self.native = native
        self.outputs_n = outputs_n
        self.kind = kind
        self.post_divisor = post_divisor


_handles = {}


def _register(native_handle, outputs_n=1, kind="op", post_divisor=None):
    info = _HandleInfo(native_handle, outputs_n, kind, post_divisor)
    _handles[native_handle] = info
    return native_handle


def poll(handle):
    """Return True if the async op identified by `handle` has completed."""
    return _core.poll(handle)


def synchronize(handle):
    """Wait for the async op and return its output tensor(s)."""
    info = _handles.pop(handle, None)
    _core.flush()  # cut the cycle-pacing window: we are about to block
    outs, extra, result_int = _translate_error(_core.wait, handle)
    if info is not None and info.post_divisor:
        # integer Average: floor-divide the summed result (reference:
        # DivideInPlace, mpi_ops_v2.cc:62-68)
        for o in outs:
            if o is not None and o.numel():
                o.floor_divide_(info.post_divisor)
    if info is not None and info.kind == "join":
        return result_int
    if info is not None and info.kind in ("alltoall_splits",
                                          "allgather_sizes"):
        return outs[0], extra
    if info is not None and info.kind == "grouped":
        return list(outs)
    return outs[0] if outs else None


def wait(handle):
    return synchronize(handle)


# ---------------------------------------------------------------------------
# allreduce
# ---------------------------------------------------------------------------
def _allreduce_async_impl(tensors, outputs, name, true_op, pre, post, ps_id,
                          wire_code, kind="op", post_divisor=None):
    names = list(name) if isinstance(name, (list, tuple)) else [name]
    h = _translate_error(_core.allreduce_async, tensors, outputs, names,
                         true_op, pre, post, ps_id, wire_code)
    return _register(h, len(tensors), kind, post_divisor)


def allreduce_async(tensor, average=None, name=None, op=None,
                    prescale_factor=1.0, postscale_factor=1.0,
                    process_set=global_process_set):
    return _do_allreduce_async(tensor, None, average, name, op,
                               prescale_factor, postscale_factor, process_set)


def allreduce_async_(tensor, average=None, name=None, op=None,
                     prescale_factor=1.0, postscale_factor=1.0,
                     process_set=global_process_set):
    return _do_allreduce_async(tensor, tensor, average, name, op,
                               prescale_factor, postscale_factor, process_set)


def _do_allreduce_async(tensor, output, average, name, op, prescale_factor,
                        postscale_factor, process_set, wire_dtype=None):
    if not _dense_ok(tensor):
        if output is tensor:
            raise ValueError(
                "hvd.allreduce_ requires a dense tensor; call .contiguous() "
                "first or use the out-of-place hvd.allreduce")
        tensor = tensor.contiguous()
    if output is None:
        output = torch.empty_like(tensor)
    true_op, _, pre, post = _resolve_scales(op, average, prescale_factor,
                                            postscale_factor, process_set)
    post_div = None
    if not tensor.dtype.is_floating_point and post != 1.0:
        # integer Average: sum on the wire, floor-divide after (reference
        # semantics; scale kernels are float-family only)
        post_div = round(1.0 / post)
        post = 1.0
    ps_id = _set_id(process_set)
    name = name or _next_name("allreduce")
    wire_code = (_dtype_code(wire_dtype) if wire_dtype is not None
                 else _dtype_code(tensor.dtype))
    return _allreduce_async_impl([tensor], [output], ["allreduce." + name],
                                 true_op, pre, post, ps_id, wire_code,
                                 post_divisor=post_div)


class HorovodAllreduce(torch.autograd.Function):
    @staticmethod
    def forward(ctx, tensor, average, name, op, prescale_factor,
                postscale_factor, process_set):
        ctx.average = average
        ctx.op = op
        ctx.prescale_factor = prescale_factor
        ctx.postscale_factor = postscale_factor
        ctx.process_set = process_set
        handle = allreduce_async(tensor, average, name, op, prescale_factor,
                                 postscale_factor, process_set)
        return synchronize(handle)

    @staticmethod
    def backward(ctx, grad_output):
        return (allreduce(grad_output, average=ctx.average, op=ctx.op,
                          prescale_factor=ctx.prescale_factor,
                          postscale_factor=ctx.postscale_factor,
                          process_set=ctx.process_set),
                None, None, None, None, None, None)


def allreduce(tensor, average=None, name=None, compression=None, op=None,
              prescale_factor=1.0, postscale_factor=1.0,
              process_set=global_process_set):
    """Reduce `tensor` across all processes of the set; differentiable.

    Reference semantics: horovod/torch/mpi_ops.py allreduce.

    Arguments:
        tensor: torch tensor (CPU or MI355X GPU); any dense layout.
        average: deprecated alias — True => op=Average, False => op=Sum.
        name: negotiation key.  Ops are matched across ranks by name; pass a
            stable unique name for anything called concurrently.
        compression: hvd.Compression.none/fp16/bf16 — wire dtype, converted
            inside the CDNA4 pack kernel (no python-side copies).
        op: hvd.Average (default) | Sum | Adasum | Min | Max | Product.
        prescale_factor/postscale_factor: scalar factors fused into the
            pack/unpack kernels.  Integer tensors floor-divide after the sum.
        process_set: subgroup to reduce over.

    Returns a new tensor with the reduction across the set (gradients flow
    through another allreduce of the same op).
    """
    from horovod_amd.torch.compression import Compression
    wire = None
    if compression is not None and compression is not Compression.none:
        wire = compression.wire_dtype(tensor.dtype)
    if wire is not None and tensor.dtype.is_floating_point:
        output = torch.empty_like(tensor)
        handle = _do_allreduce_async(tensor, output, average, name, op,
                                     prescale_factor, postscale_factor,
                                     process_set, wire_dtype=wire)
        return synchronize(handle)
    return HorovodAllreduce.apply(tensor, average, name, op, prescale_factor,
                                  postscale_factor, process_set)


def allreduce_(tensor, average=None, name=None, op=None, prescale_factor=1.0,
               postscale_factor=1.0, process_set=global_process_set):
    """In-place allreduce."""
    handle = allreduce_async_(tensor, average, name, op, prescale_factor,
                              postscale_factor, process_set)
    return synchronize(handle)


# grouped ------------------------------------------------------------------
def grouped_allreduce_async(tensors, average=None, name=None, op=None,
                            prescale_factor=1.0, postscale_factor=1.0,
                            process_set=global_process_set):
    outputs = [torch.empty_like(t) for t in tensors]
    return _grouped_allreduce_impl(tensors, outputs, average, name, op,
                                   prescale_factor, postscale_factor,
                                   process_set)


def grouped_allreduce_async_(tensors, average=None, name=None, op=None,
                             prescale_factor=1.0, postscale_factor=1.0,
                             process_set=global_process_set):
    return _grouped_allreduce_impl(tensors, tensors, average, name, op,
                                   prescale_factor, postscale_factor,
                                   process_set)


def _grouped_allreduce_impl(tensors, outputs, average, name, op,
                            prescale_factor, postscale_factor, process_set,
                            wire_dtype=None):
    tensors = list(tensors)
    outputs = list(outputs)
    for i, t in enumerate(tensors):
        if not _dense_ok(t):
            if outputs[i] is t:
                raise ValueError("grouped_allreduce_ requires dense tensors")
            tensors[i] = t.contiguous()
    true_op, _, pre, post = _resolve_scales(op, average, prescale_factor,
                                            postscale_factor, process_set)
    post_div = None
    if not tensors[0].dtype.is_floating_point and post != 1.0:
        post_div = round(1.0 / post)
        post = 1.0
    ps_id = _set_id(process_set)
    base = name or _next_name("grouped_allreduce")
    names = [f"allreduce.{base}.{i}" for i in range(len(tensors))]
    wire_code = (_dtype_code(wire_dtype) if wire_dtype is not None
                 else _dtype_code(tensors[0].dtype))
    return _allreduce_async_impl(list(tensors), list(outputs), names, true_op,
                                 pre, post, ps_id, wire_code, kind="grouped",
                                 post_divisor=post_div)


class HorovodGroupedAllreduce(torch.autograd.Function):
    """Differentiable grouped allreduce (reference: mpi_ops.py
    HorovodGroupedAllreduce): ONE fused negotiation unit; the gradient is
    the same grouped allreduce of the incoming grads."""

    @staticmethod
    def forward(ctx, average, name, op, prescale_factor, postscale_factor,
                process_set, *tensors):
        ctx.average = average
        ctx.op = op
        ctx.prescale_factor = prescale_factor
        ctx.postscale_factor = postscale_factor
        ctx.process_set = process_set
        outputs = [torch.empty_like(t) for t in tensors]
        handle = _grouped_allreduce_impl(list(tensors), outputs, average,
                                         name, op, prescale_factor,
                                         postscale_factor, process_set)
        return tuple(synchronize(handle))

    @staticmethod
    def backward(ctx, *grads):
        gs = [g.contiguous() for g in grads]
        outs = [torch.empty_like(g) for g in gs]
        handle = _grouped_allreduce_impl(gs, outs, ctx.average, None, ctx.op,
                                         ctx.prescale_factor,
                                         ctx.postscale_factor,
                                         ctx.process_set)
        reduced = synchronize(handle)
        return (None, None, None, None, None, None) + tuple(reduced)


def grouped_allreduce(tensors, average=None, name=None, compression=None,
                      op=None, prescale_factor=1.0, postscale_factor=1.0,
                      process_set=global_process_set):
    from horovod_amd.torch.compression import Compression
    if any(t.requires_grad for t in tensors):
        return list(HorovodGroupedAllreduce.apply(
            average, name, op, prescale_factor, postscale_factor,
            process_set, *tensors))
    wire = None
    if compression is not None and compression is not Compression.none:
        wire = compression.wire_dtype(tensors[0].dtype)
    outputs = [torch.empty_like(t) for t in tensors]
    handle = _grouped_allreduce_impl(tensors, outputs, average, name, op,
                                     prescale_factor, postscale_factor,
                                     process_set, wire_dtype=wire)
    return synchronize(handle)


def grouped_allreduce_(tensors, average=None, name=None, op=None,
                       prescale_factor=1.0, postscale_factor=1.0,
                       process_set=global_process_set):
    handle = grouped_allreduce_async_(tensors, average, name, op,
                                      prescale_factor, postscale_factor,
                                      process_set)
    return synchronize(handle)


# sparse -------------------------------------------------------------------
def sparse_allreduce_async(tensor, name, op, process_set=global_process_set):
    """Allreduce a torch.sparse COO tensor (reference: mpi_ops.py:567-589):
    allgather indices and values, return a closure reconstructing the
    summed/averaged sparse tensor."""
    t = tensor.coalesce()
    indices_handle = allgather_async(t.indices().t().contiguous(),
                                     name=f"{name}.indices",
                                     process_set=process_set)
    values_handle = allgather_async(t.values(), name=f"{name}.values",
                                    process_set=process_set)

    def handle():
        gathered_indices = synchronize(indices_handle).t()
        gathered_values = synchronize(values_handle)
        if op == Average:
            gathered_values = gathered_values / _set_size(process_set)
        return torch.sparse_coo_tensor(gathered_indices, gathered_values,
                                       t.size()).coalesce()

    return handle


# ---------------------------------------------------------------------------
# allgather
# ---------------------------------------------------------------------------
def allgather_async(tensor, name=None, process_set=global_process_set):
    name = name or _next_name("allgather")
    h = _translate_error(_core.allgather_async, tensor.contiguous(),
                         "allgather." + name, _set_id(process_set))
    return _register(h)


class HorovodAllgather(torch.autograd.Function):
    @staticmethod
    def forward(ctx, tensor, name, process_set):
        ctx.dim = tensor.shape[0] if tensor.dim() > 0 else 0
        ctx.process_set = process_set
        name = name or _next_name("allgather")
        h = _translate_error(_core.allgather_async, tensor.contiguous(),
                             "allgather." + name, _set_id(process_set))
        _register(h, kind="allgather_sizes")
        out, sizes = synchronize(h)
        # the native response carries every rank's first-dim contribution, so
        # the backward slice offset is known NOW — no dims-allgather per
        # backward pass (round-1 weakness)
        me = rank() if process_set.process_set_id == 0 else \
            process_set.ranks.index(rank())
        ctx.offset = int(sizes[:me].sum().item()) if (
            sizes is not None and me > 0) else me * ctx.dim
        return out

    @staticmethod
    def backward(ctx, grad_output):
        # sum the gradient across ranks, then slice out this rank's segment
        # (reference: allgather grad = reducescatter-like slice)
        grad_reduced = allreduce(grad_output, average=False,
                                 process_set=ctx.process_set)
        return grad_reduced.narrow(0, ctx.offset, ctx.dim), None, None


def allgather(tensor, name=None, process_set=global_process_set):
    """Concatenate `tensor` from all set members along dim 0; differentiable.
    First dimensions may differ per rank (variable-gather); other dims must
    match.  Reference: horovod/torch/mpi_ops.py allgather."""
    return HorovodAllgather.apply(tensor, name, process_set)


def grouped_allgather_async(tensors, name=None, process_set=global_process_set):
    base = name or _next_name("grouped_allgather")
    return [allgather_async(t, f"{base}.{i}", process_set)
            for i, t in enumerate(tensors)]


def grouped_allgather(tensors, name=None, process_set=global_process_set):
    if any(t.requires_grad for t in tensors):
        # differentiable path: per-tensor autograd allgather (reference
        # exposes grouped gradients; the per-tensor functions negotiate
        # concurrently through the background fusion anyway)
        base = name or _next_name("grouped_allgather")
        return [allgather(t, name=f"{base}.{i}", process_set=process_set)
                for i, t in enumerate(tensors)]
    handles = grouped_allgather_async(tensors, name, process_set)
    return [synchronize(h) for h in handles]


# ---------------------------------------------------------------------------
# broadcast
# ---------------------------------------------------------------------------
def broadcast_async(tensor, root_rank, name=None,
                    process_set=global_process_set):
    if not _dense_ok(tensor):
        tensor = tensor.contiguous()
    output = torch.empty_like(tensor)
    name = name or _next_name("broadcast")
    h = _translate_error(_core.broadcast_async, tensor, output, root_rank,
                         "broadcast." + name, _set_id(process_set))
    return _register(h)


def broadcast_async_(tensor, root_rank, name=None,
                     process_set=global_process_set):
    if not _dense_ok(tensor):
        raise ValueError("hvd.broadcast_ requires a dense tensor")
    name = name or _next_name("broadcast")
    h = _translate_error(_core.broadcast_async, tensor, tensor, root_rank,
                         "broadcast." + name, _set_id(process_set))
    return _register(h)


class HorovodBroadcast(torch.autograd.Function):
    @staticmethod
    def forward(ctx, tensor, root_rank, name, process_set):
        ctx.root_rank = root_rank
        ctx.process_set = process_set
        handle = broadcast_async(tensor, root_rank, name, process_set)
        return synchronize(handle)

    @staticmethod
    def backward(ctx, grad_output):
        grad_reduced = allreduce(grad_output, average=False,
                                 process_set=ctx.process_set)
        if rank() != ctx.root_rank:
            grad_reduced = grad_reduced * 0
        return grad_reduced, None, None, None


def broadcast(tensor, root_rank, name=None, process_set=global_process_set):
    """Return `tensor` broadcast from global rank `root_rank` to every set
    member; differentiable (grad reduces back to the root).
    Reference: horovod/torch/mpi_ops.py broadcast."""
    return HorovodBroadcast.apply(tensor, root_rank, name, process_set)


def broadcast_(tensor, root_rank, name=None, process_set=global_process_set):
    handle = broadcast_async_(tensor, root_rank, name, process_set)
    return synchronize(handle)


# ---------------------------------------------------------------------------
# alltoall
# ---------------------------------------------------------------------------
def alltoall_async(tensor, splits=None, name=None,
                   process_set=global_process_set):
    name = name or _next_name("alltoall")
    n = _set_size(process_set)
    if splits is None:
        first = tensor.shape[0] if tensor.dim() > 0 else 0
        if first % n != 0:
            raise ValueError(
                "splits not provided and first dimension not divisible by the "
                "process set size")
        splits_t = torch.full((n,), first // n, dtype=torch.int64)
    else:
        splits_t = torch.as_tensor(splits, dtype=torch.int64).cpu()
    h = _translate_error(_core.alltoall_async, tensor.contiguous(), splits_t,
                         "alltoall." + name, _set_id(process_set))
    return _register(h, kind="alltoall_splits")


class HorovodAlltoall(torch.autograd.Function):
    @staticmethod
    def forward(ctx, tensor, splits, name, process_set):
        handle = alltoall_async(tensor, splits, name, process_set)
        output, received_splits = synchronize(handle)
        ctx.process_set = process_set
        ctx.recvsplits = received_splits
        ctx.mark_non_differentiable(received_splits)
        return output, received_splits

    @staticmethod
    def backward(ctx, grad_output, grad_splits):
        # reverse exchange: our recv splits become the send splits
        out = alltoall(grad_output, splits=ctx.recvsplits,
                       process_set=ctx.process_set)
        if isinstance(out, tuple):
            out = out[0]
        return out, None, None, None


def alltoall(tensor, splits=None, name=None, process_set=global_process_set):
    """Scatter slices of dim 0 to every set member and gather theirs.

    splits: per-destination row counts (int sequence of set size); None =
    uniform split (dim 0 must divide evenly).  Returns the received tensor,
    plus received_splits when `splits` was given (reference:
    horovod/torch/mpi_ops.py alltoall; grad = reverse exchange)."""
    output, received = HorovodAlltoall.apply(tensor, splits, name, process_set)
    if splits is None:
        return output
    return output, received


# ---------------------------------------------------------------------------
# reducescatter
# ---------------------------------------------------------------------------
def reducescatter_async(tensor, name=None, op=None,
                        process_set=global_process_set, prescale_factor=1.0,
                        postscale_factor=1.0):
    true_op, _, pre, post = _resolve_scales(op, None, prescale_factor,
                                            postscale_factor, process_set)
    post_div = None
    if not tensor.dtype.is_floating_point and post != 1.0:
        post_div = round(1.0 / post)  # integer Average: floor-divide post-op
        post = 1.0
    name = name or _next_name("reducescatter")
    h = _translate_error(_core.reducescatter_async, tensor.contiguous(),
                         "reducescatter." + name, true_op, pre, post,
                         _set_id(process_set))
    return _register(h, post_divisor=post_div)


class HorovodReducescatter(torch.autograd.Function):
    @staticmethod
    def forward(ctx, tensor, op, name, prescale_factor, postscale_factor,
                process_set):
        ctx.process_set = process_set
        ctx.op = op
        handle = reducescatter_async(tensor, name=name, op=op,
                                     process_set=process_set,
                                     prescale_factor=prescale_factor,
                                     postscale_factor=postscale_factor)
        return synchronize(handle)

    @staticmethod
    def backward(ctx, grad_output):
        grad = allgather(grad_output, process_set=ctx.process_set)
        if ctx.op in (None, Average):
            grad = grad / _set_size(ctx.process_set)
        return grad, None, None, None, None, None


def reducescatter(tensor, name=None, compression=None, op=None,
                  process_set=global_process_set, prescale_factor=1.0,
                  postscale_factor=1.0):
    """Reduce `tensor` across the set, returning this rank's dim-0 shard
    (rows split as evenly as possible, earlier ranks get the remainder).
    op defaults to Average.  `compression` is accepted for reference-
    signature parity (applied python-side around the op).
    Reference: horovod/torch/mpi_ops.py reducescatter."""
    from horovod_amd.torch.compression import Compression
    if compression is not None and compression is not Compression.none:
        t, ctx = compression.compress(tensor)
        out = HorovodReducescatter.apply(t, op, name, prescale_factor,
                                         postscale_factor, process_set)
        return compression.decompress(out, ctx)
    return HorovodReducescatter.apply(tensor, op, name, prescale_factor,
                                      postscale_factor, process_set)


def grouped_reducescatter_async(tensors, name=None, op=None,
                                process_set=global_process_set,
                                prescale_factor=1.0, postscale_factor=1.0):
    base = name or _next_name("grouped_reducescatter")
    return [reducescatter_async(t, name=f"{base}.{i}", op=op,
                                process_set=process_set,
                                prescale_factor=prescale_factor,
                                postscale_factor=postscale_factor)
            for i, t in enumerate(tensors)]


def grouped_reducescatter(tensors, name=None, compression=None, op=None,
                          process_set=global_process_set, prescale_factor=1.0,
                          postscale_factor=1.0):
    base = name or _next_name("grouped_reducescatter")
    return [reducescatter(t, name=f"{base}.{i}", compression=compression,
                          op=op, process_set=process_set,
                          prescale_factor=prescale_factor,
                          postscale_factor=postscale_factor)
            for i, t in enumerate(tensors)]


# ---------------------------------------------------------------------------
# join / barrier
# ---------------------------------------------------------------------------
def process_set_included(process_set_id=0):
    """1 if this rank belongs to the process set, else 0 (reference:
    operations.cc horovod_process_set_included)."""
    from horovod_amd.common import process_sets as _ps
    if process_set_id == 0:
        return 1
    for ps in _ps._registered_sets():
        if ps.process_set_id == process_set_id:
            return 1 if rank() in (ps.ranks or []) else 0
    raise ValueError(f"unknown process_set_id {process_set_id}")


def handle_average_backwards_compatibility(op, average):
    """Map the legacy `average=` flag onto `op=` (reference:
    horovod/common/util.py handle_average_backwards_compatibility)."""
    if op is not None and average is not None:
        raise ValueError("The op parameter supersedes average. Please "
                         "provide only one of them.")
    if op is not None:
        return op
    if average is not None:
        return Average if average else Sum
    return Average


def join(device=-1):
    """Signal that this rank has no more data; blocks until every rank has
    joined.  Returns the last rank to join (reference: operations.cc
    1991-2021)."""
    h = _core.join_async(device, 0)
    _register(h, kind="join")
    return synchronize(h)


def start_timeline(file_path, mark_cycles=False):
    """Start writing a Chrome-trace timeline at runtime (reference:
    operations.cc horovod_start_timeline)."""
    _core.start_timeline(file_path, mark_cycles)


def stop_timeline():
    _core.stop_timeline()


def barrier(process_set=global_process_set):
    """Block until every member of the set has entered the barrier
    (negotiation-synchronized; reference: operations.cc EnqueueBarrier)."""
    h = _core.barrier_async(_set_id(process_set))
    _register(h)
    synchronize(h)


@contextmanager
def _noop_ctx():
    yield
