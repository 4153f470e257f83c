from horovod_amd.runner.launch import main, run_command_local  # noqa: F401
