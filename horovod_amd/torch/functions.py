"""State synchronization helpers.

Reference: horovod/torch/functions.py:30-279 — broadcast_parameters,
broadcast_optimizer_state, broadcast_object, allgather_object.
"""
import io

import cloudpickle
import torch

from horovod_amd.common.process_sets import global_process_set
from horovod_amd.torch.mpi_ops import (allgather, broadcast_,
                                       broadcast_async_, synchronize, rank)


def broadcast_parameters(params, root_rank, process_set=global_process_set,
                         prefix="Parameter"):
    """Broadcast parameters from root_rank to all other processes
    (reference: functions.py:30-72)."""
    if isinstance(params, dict):
        params = sorted(params.items())
    elif isinstance(params, list):
        params = [(str(i), p) for i, p in enumerate(params)]
    else:
        raise ValueError("invalid params of type: %s" % type(params))

    handles = []
    for name, p in params:
        if p is None:
            continue
        handles.append(broadcast_async_(p.data if hasattr(p, "data") else p,
                                        root_rank,
                                        name=f"{prefix}.{name}",
                                        process_set=process_set))
    for h in handles:
        synchronize(h)


def broadcast_object(obj, root_rank=0, name=None,
                     process_set=global_process_set):
    """Serialize (cloudpickle) + broadcast an arbitrary object
    (reference: functions.py:201-246)."""
    name = name or "broadcast_object"
    if rank() == root_rank:
        b = io.BytesIO()
        cloudpickle.dump(obj, b)
        payload = torch.ByteTensor(bytearray(b.getvalue()))
        sz = torch.IntTensor([payload.numel()])
        broadcast_(sz, root_rank, name=f"{name}.sz", process_set=process_set)
    else:
        sz = torch.IntTensor([0])
        broadcast_(sz, root_rank, name=f"{name}.sz", process_set=process_set)
        payload = torch.ByteTensor(int(sz.item()))
    broadcast_(payload, root_rank, name=f"{name}.data", process_set=process_set)
    if rank() != root_rank:
        obj = cloudpickle.load(io.BytesIO(payload.numpy().tobytes()))
    return obj


def allgather_object(obj, name=None, process_set=global_process_set):
    """Allgather arbitrary picklable objects; returns a list by set-local
    rank (reference: functions.py:248-279)."""
    name = name or "allgather_object"
    b = io.BytesIO()
    cloudpickle.dump(obj, b)
    payload = torch.ByteTensor(bytearray(b.getvalue()))
    sizes = allgather(torch.IntTensor([payload.numel()]),
                      name=f"{name}.sz", process_set=process_set)
    gathered = allgather(payload, name=f"{name}.data", process_set=process_set)
    out, off = [], 0
    for s in sizes.tolist():
        out.append(cloudpickle.loads(
            gathered[off:off + s].numpy().tobytes()))
        off += s
    return out


def broadcast_optimizer_state(optimizer, root_rank, model=None,
                              process_set=global_process_set):
    """Broadcast an optimizer's state from root (reference:
    functions.py:74-199).  Tensor state entries are broadcast in place;
    non-tensor entries (step counters, hyperparameters) travel via
    broadcast_object."""
    if isinstance(optimizer, torch.optim.LBFGS):
        raise ValueError("cannot broadcast torch.optim.LBFGS state")

    state_dict = optimizer.state_dict()

    # identify sparse parameters via the model (reference functions.py:81-88):
    # sparse-Embedding params need sparse dummy grads during state init or the
    # optimizer builds the wrong state structure
    sparse_ids = set()
    if model is not None:
        for m in model.modules():
            if isinstance(m, torch.nn.Embedding) and m.sparse:
                for p in m.parameters():
                    sparse_ids.add(id(p))

    # Newly created optimizers have no state; initialize it with a zero-grad
    # step on EVERY rank so the structure is identical before broadcasting
    # (reference functions.py:90-109).
    if len(state_dict["state"]) == 0:
        for group in optimizer.param_groups:
            for p in group["params"]:
                if p.requires_grad:
                    p.grad = p.data.new_zeros(p.size())
                    if (isinstance(optimizer, torch.optim.SparseAdam)
                            or id(p) in sparse_ids):
                        p.grad = p.grad.to_sparse()
        # call the WRAPPED optimizer's step to avoid firing allreduce hooks
        if hasattr(optimizer, "reset_distributed_state"):
            super(optimizer.__class__, optimizer).step()
        else:
            optimizer.step()
        state_dict = optimizer.state_dict()
        optimizer.zero_grad(set_to_none=True)
    if len(state_dict["state"]) == 0:
        return  # stateless optimizer (plain SGD without momentum)

    # ensure every rank has state initialized with the same structure: on
    # root, missing state stays; on workers we rebuild from root's metadata.
    meta = None
    if rank() == root_rank:
        meta = {
            "param_groups": state_dict["param_groups"],
            "state_keys": {
                pid: [(k, (tuple(v.shape), str(v.dtype)) if torch.is_tensor(v)
                       else ("scalar", v))
                      for k, v in s.items()]
                for pid, s in state_dict["state"].items()
            },
        }
    meta = broadcast_object(meta, root_rank, name="opt_state_meta",
                            process_set=process_set)

    # map param id -> actual parameter tensor for shape/device/dtype
    id_to_param = {}
    idx = 0
    for group in optimizer.param_groups:
        for p in group["params"]:
            id_to_param[idx] = p
            idx += 1

    # rebuild non-root state skeleton, then broadcast each tensor in place
    handles = []
    scalars = {}
    for pid, keys in meta["state_keys"].items():
        pid = int(pid)
        for key, desc in keys:
            if desc[0] == "scalar":
                scalars[f"{pid}.{key}"] = desc[1]
                continue
            shape, dtype_s = desc
            dtype = getattr(torch, dtype_s.replace("torch.", ""))
            if rank() == root_rank:
                t = state_dict["state"][pid][key]
            else:
                ref = id_to_param.get(pid)
                device = ref.device if ref is not None else "cpu"
                t = torch.zeros(shape, dtype=dtype, device=device)
                optimizer.state.setdefault(ref, {})[key] = t
            handles.append(broadcast_async_(t, root_rank,
                                            name=f"opt_state.{pid}.{key}",
                                            process_set=process_set))
    for h in handles:
        synchronize(h)

    # non-tensor scalar state + param group hyperparameters
    scalars = broadcast_object(scalars, root_rank, name="opt_state_scalars",
                               process_set=process_set)
    if rank() != root_rank:
        for key, val in scalars.items():
            pid_s, k = key.split(".", 1)
            ref = id_to_param.get(int(pid_s))
            if ref is not None:
                optimizer.state.setdefault(ref, {})[k] = val
        for g, meta_g in zip(optimizer.param_groups, meta["param_groups"]):
            for k, v in meta_g.items():
                if k != "params":
                    g[k] = v
