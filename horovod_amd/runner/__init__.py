"""Launcher package (reference: horovod/runner).  Imports are lazy so
`python -m horovod_amd.runner.launch` doesn't double-import the module."""


def main(argv=None):
    from horovod_amd.runner.launch import main as _main
    return _main(argv)


def run_command_local(*a, **kw):
    from horovod_amd.runner.launch import run_command_local as _r
    return _r(*a, **kw)
