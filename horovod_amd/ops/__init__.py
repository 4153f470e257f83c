from horovod_amd.ops.fused_sgd import FusedSGD  # noqa: F401
from horovod_amd.ops.fused_adamw import FusedAdamW  # noqa: F401
from horovod_amd.ops.fused_bn import (FusedBNAddReLU,  # noqa: F401
                                      FusedBNReLU)
