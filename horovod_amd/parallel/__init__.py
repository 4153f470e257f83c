from horovod_amd.parallel.groups import grid_process_sets  # noqa: F401
