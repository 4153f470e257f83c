"""TorchState: elastic state handlers for torch objects.

Reference: horovod/torch/elastic/state.py:27-135 (TorchState +
Model/Optimizer/Sampler handlers).
"""
import copy

import torch

from horovod_amd.common.elastic import ObjectState
from horovod_amd.torch.functions import (broadcast_object,
                                         broadcast_optimizer_state,
                                         broadcast_parameters)
from horovod_amd.torch.mpi_ops import rank


class StateHandler:
    """Base class for custom state handlers (reference: state.py
    StateHandler).  Register a type with set_handler_registry/
    get_handler_registry so TorchState picks it up for your objects."""

    def __init__(self, value):
        self.value = value

    def save(self):
        raise NotImplementedError()

    def restore(self):
        raise NotImplementedError()

    def sync(self):
        raise NotImplementedError()

    def set_value(self, value):
        self.value = value
        self.save()


class ModelStateHandler(StateHandler):
    def __init__(self, model):
        super().__init__(model)
        self._saved_model_state = copy.deepcopy(model.state_dict())

    def save(self):
        self._saved_model_state = copy.deepcopy(self.value.state_dict())

    def restore(self):
        self.value.load_state_dict(self._saved_model_state)

    def sync(self):
        broadcast_parameters(self.value.state_dict(), root_rank=0)


class OptimizerStateHandler(StateHandler):
    def __init__(self, optimizer):
        super().__init__(optimizer)
        self._saved_state = copy.deepcopy(optimizer.state_dict())

    def save(self):
        self._saved_state = copy.deepcopy(self.value.state_dict())

    def restore(self):
        try:
            self.value.load_state_dict(self._saved_state)
        except ValueError:
            pass  # param groups changed shape; keep current
        if hasattr(self.value, "reset_distributed_state"):
            self.value.reset_distributed_state()

    def sync(self):
        if hasattr(self.value, "reset_distributed_state"):
            self.value.reset_distributed_state()
        broadcast_optimizer_state(self.value, root_rank=0)


class SamplerStateHandler(StateHandler):
    def save(self):
        self._saved = self.value.state_dict()

    def restore(self):
        self.value.load_state_dict(self._saved)

    def sync(self):
        state = broadcast_object(self.value.state_dict()
                                 if rank() == 0 else None, root_rank=0,
                                 name="sampler_state")
        self.value.load_state_dict(state)
        # reshard remaining work over the new world size
        self.value.reset()


def _default_registry():
    from horovod_amd.torch.elastic.sampler import ElasticSampler
    return [(torch.nn.Module, ModelStateHandler),
            (torch.optim.Optimizer, OptimizerStateHandler),
            (ElasticSampler, SamplerStateHandler)]


_handler_registry = None


def get_handler_registry():
    """The (type, StateHandler) pairs TorchState consults, most specific
    first (reference: state.py get_handler_registry)."""
    global _handler_registry
    if _handler_registry is None:
        _handler_registry = _default_registry()
    return list(_handler_registry)


def set_handler_registry(registry):
    """Replace the handler registry (reference: set_handler_registry) —
    e.g. to add a handler for a custom scheduler type."""
    global _handler_registry
    _handler_registry = list(registry)


def _get_handler(v):
    for typ, handler_cls in get_handler_registry():
        if isinstance(v, typ):
            return handler_cls(v)
    return None


class TorchState(ObjectState):
    """State of a torch training job: models, optimizers, samplers + plain
    attributes (epoch, batch, ...)."""

    def __init__(self, model=None, optimizer=None, **kwargs):
        kwargs.update(dict(model=model, optimizer=optimizer))
        self._handlers = {}
        plain = {}
        for name, value in list(kwargs.items()):
            if value is None:
                continue
            h = _get_handler(value)
            if h is not None:
                self._handlers[name] = h
                setattr(self, name, value)
            else:
                plain[name] = value
        super().__init__(bcast_object=broadcast_object,
                         get_rank=rank, **plain)

    def save(self):
        for h in self._handlers.values():
            h.save()
        super().save()

    def restore(self):
        for h in self._handlers.values():
            h.restore()
        super().restore()

    def sync(self):
        for h in self._handlers.values():
            h.sync()
        super().sync()

    def __setattr__(self, name, value):
        if hasattr(self, "_handlers") and name in self._handlers:
            self._handlers[name].set_value(value)
        super().__setattr__(name, value)
