from horovod_amd.ops.fused_sgd import FusedSGD  # noqa: F401
