from horovod_amd.torch.elastic.state import (TorchState,  # noqa: F401
                                              StateHandler,
                                              ModelStateHandler,
                                              OptimizerStateHandler,
                                              SamplerStateHandler,
                                              get_handler_registry,
                                              set_handler_registry)
from horovod_amd.torch.elastic.sampler import ElasticSampler  # noqa: F401

from horovod_amd.common.elastic import (ObjectState, State,  # noqa: F401
                                        run_fn, _rendezvous_reset)


def run(func):
    """Elastic training decorator (reference: torch/elastic/__init__.py:23):

        @hvd.elastic.run
        def train(state): ...
        state = hvd.elastic.TorchState(model, optimizer, epoch=0, batch=0)
        train(state)
    """
    return run_fn(func, _rendezvous_reset)
