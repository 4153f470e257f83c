"""`hvdrun` — the horovodrun-compatible launcher.

Reference: horovod/runner/launch.py (argparse surface) + gloo_run.py (slot
env protocol, ssh fan-out).  The MI355X deployment target is one node with
8 GPUs, so the primary path is local exec with the per-slot environment;
multi-host uses ssh fan-out with the same env protocol.  No MPI.
"""
import argparse
import os
import shlex
import socket
import subprocess
import sys


def find_free_port():
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def parse_host_spec(hosts, np):
    """'host1:4,host2:4' -> [(host, slots), ...]; None -> localhost:np."""
    if not hosts:
        return [("127.0.0.1", np)]
    out = []
    for part in hosts.split(","):
        if ":" in part:
            h, s = part.rsplit(":", 1)
            out.append((h, int(s)))
        else:
            out.append((part, 1))
    return out


def slot_env(rank, size, local_rank, local_size, cross_rank, cross_size,
             addr, port, base_env=None):
    env = dict(base_env or os.environ)
    env.update({
        "HOROVOD_RANK": str(rank),
        "HOROVOD_SIZE": str(size),
        "HOROVOD_LOCAL_RANK": str(local_rank),
        "HOROVOD_LOCAL_SIZE": str(local_size),
        "HOROVOD_CROSS_RANK": str(cross_rank),
        "HOROVOD_CROSS_SIZE": str(cross_size),
        "HOROVOD_CONTROLLER_ADDR": addr,
        "HOROVOD_CONTROLLER_PORT": str(port),
        # keep dmabuf IPC for RCCL (see environment notes)
        "HSA_ENABLE_IPC_MODE_LEGACY":
            env.get("HSA_ENABLE_IPC_MODE_LEGACY", "0"),
    })
    return env


def run_command_local(np, command, env=None, port=None, verbose=False,
                      stdout=None):
    """Launch `command` (list) np times on this machine with the slot env
    protocol.  Returns list of exit codes."""
    port = port or find_free_port()
    procs = []
    for rank in range(np):
        e = slot_env(rank, np, rank, np, 0, 1, "127.0.0.1", port, env)
        if verbose:
            print(f"[hvdrun] starting rank {rank}: {' '.join(command)}")
        procs.append(subprocess.Popen(command, env=e, stdout=stdout,
                                      stderr=subprocess.STDOUT
                                      if stdout else None))
    codes = [p.wait() for p in procs]
    return codes


def _ssh_command(host, command, env_vars):
    exports = " ".join(f"{k}={shlex.quote(v)}" for k, v in env_vars.items())
    return ["ssh", "-o", "StrictHostKeyChecking=no", host,
            f"cd {shlex.quote(os.getcwd())} && env {exports} "
            f"{' '.join(shlex.quote(c) for c in command)}"]


def run_distributed(np, hosts, command, env=None, port=None, verbose=False):
    """Multi-host launch over ssh (reference: gloo_run.py:242-303)."""
    alloc = parse_host_spec(hosts, np)
    total = sum(s for _, s in alloc)
    if total < np:
        raise ValueError(f"host slots ({total}) < np ({np})")
    port = port or find_free_port()
    addr = alloc[0][0]
    procs = []
    rank = 0
    for cross_rank, (host, slots) in enumerate(alloc):
        for local_rank in range(slots):
            if rank >= np:
                break
            e = slot_env(rank, np, local_rank, min(slots, np - rank + local_rank),
                         cross_rank, len(alloc), addr, port, env)
            local = host in ("127.0.0.1", "localhost",
                             socket.gethostname())
            if local:
                procs.append(subprocess.Popen(command, env=e))
            else:
                env_vars = {k: v for k, v in e.items()
                            if k.startswith(("HOROVOD_", "NCCL_", "RCCL_",
                                             "HSA_", "PATH", "PYTHONPATH"))}
                procs.append(subprocess.Popen(_ssh_command(host, command,
                                                           env_vars)))
            rank += 1
    codes = [p.wait() for p in procs]
    return codes


def build_parser():
    parser = argparse.ArgumentParser(
        prog="hvdrun",
        description="Launch a horovod_amd training job "
                    "(horovodrun-compatible).")
    parser.add_argument("-np", "--num-proc", type=int, required=False,
                        default=1, help="number of processes")
    parser.add_argument("-H", "--hosts", default=None,
                        help="host1:slots,host2:slots")
    parser.add_argument("--hostfile", default=None,
                        help="file with 'host slots=N' lines")
    parser.add_argument("--gloo", action="store_true",
                        help="accepted for compatibility (always TCP/gloo-"
                             "style here)")
    parser.add_argument("--mpi", action="store_true",
                        help="accepted for compatibility; ignored (no MPI)")
    parser.add_argument("--verbose", action="store_true")
    parser.add_argument("--start-timeout", type=int, default=600)
    parser.add_argument("--fusion-threshold-mb", type=float, default=None)
    parser.add_argument("--cycle-time-ms", type=float, default=None)
    parser.add_argument("--cache-capacity", type=int, default=None)
    parser.add_argument("--timeline-filename", default=None)
    parser.add_argument("--autotune", action="store_true")
    parser.add_argument("--network-interface", default=None,
                        help="NIC(s) for RCCL bootstrap "
                             "(sets NCCL_SOCKET_IFNAME)")
    parser.add_argument("--log-level", default=None,
                        choices=["trace", "debug", "info", "warning",
                                 "error", "fatal"])
    parser.add_argument("--min-np", type=int, default=None,
                        help="elastic: minimum np")
    parser.add_argument("--max-np", type=int, default=None,
                        help="elastic: maximum np")
    parser.add_argument("--host-discovery-script", default=None,
                        help="elastic: executable printing host:slots lines")
    parser.add_argument("-cb", "--check-build", action="store_true",
                        help="print available features/frameworks and exit "
                             "(reference: horovodrun --check-build)")
    parser.add_argument("--config-file", default=None,
                        help="YAML file with launcher settings (reference: "
                             "launch.py:581-585)")
    parser.add_argument("command", nargs=argparse.REMAINDER,
                        help="training command")
    return parser


def apply_config_file(args):
    if not args.config_file:
        return args
    import yaml
    with open(args.config_file) as f:
        cfg = yaml.safe_load(f) or {}
    for key, val in cfg.items():
        attr = key.replace("-", "_")
        if hasattr(args, attr) and getattr(args, attr) in (None, False):
            setattr(args, attr, val)
    return args


def check_build():
    print("""horovod_amd (MI355X-native)

Available frameworks:
    [X] PyTorch (ROCm)
    [ ] TensorFlow (out of scope: single-stack design)
    [ ] MXNet (out of scope)

Available controllers:
    [X] TCP star (gloo-equivalent; built in)
    [ ] MPI (replaced by design)

Available tensor operations:
    [X] RCCL over xGMI (GPU)
    [X] TCP star (CPU)
    [X] CDNA4 fused kernels: pack/scale/convert, Adasum, SGD, BN+ReLU""")
    return 0


def main(argv=None):
    args = build_parser().parse_args(argv)
    args = apply_config_file(args)
    if args.check_build:
        return check_build()
    if not args.command:
        print("hvdrun: no command given", file=sys.stderr)
        return 1
    command = args.command
    if command and command[0] == "--":
        command = command[1:]
    env = dict(os.environ)
    if args.fusion_threshold_mb is not None:
        env["HOROVOD_FUSION_THRESHOLD"] = str(
            int(args.fusion_threshold_mb * 1024 * 1024))
    if args.cycle_time_ms is not None:
        env["HOROVOD_CYCLE_TIME"] = str(args.cycle_time_ms)
    if args.cache_capacity is not None:
        env["HOROVOD_CACHE_CAPACITY"] = str(args.cache_capacity)
    if args.timeline_filename:
        env["HOROVOD_TIMELINE"] = args.timeline_filename
    if args.autotune:
        env["HOROVOD_AUTOTUNE"] = "1"
    if args.network_interface:
        env["NCCL_SOCKET_IFNAME"] = args.network_interface
    if args.log_level:
        env["HOROVOD_LOG_LEVEL"] = args.log_level
    if args.start_timeout:
        env["HOROVOD_START_TIMEOUT"] = str(args.start_timeout)

    hosts = args.hosts
    if args.hostfile:
        specs = []
        with open(args.hostfile) as f:
            for line in f:
                line = line.strip()
                if not line or line.startswith("#"):
                    continue
                parts = line.split()
                host = parts[0]
                slots = 1
                for p in parts[1:]:
                    if p.startswith("slots="):
                        slots = int(p.split("=", 1)[1])
                specs.append(f"{host}:{slots}")
        hosts = ",".join(specs)

    if args.host_discovery_script:
        from horovod_amd.runner.elastic_driver import run_elastic
        return run_elastic(args, command, env)

    # NIC auto-discovery for multi-host launches (reference:
    # driver_service.py probes common interfaces before launch); an explicit
    # --network-interface always wins
    if hosts and "NCCL_SOCKET_IFNAME" not in env:
        from horovod_amd.runner.network import resolve_nics
        host_names = [h.split(":")[0] for h in hosts.split(",")]
        nics = resolve_nics(args.network_interface, host_names,
                            verbose=args.verbose)
        if nics:
            env["NCCL_SOCKET_IFNAME"] = nics
            if args.verbose:
                print(f"[hvdrun] common NICs: {nics}")

    if hosts:
        codes = run_distributed(args.num_proc, hosts, command, env=env,
                                verbose=args.verbose)
    else:
        codes = run_command_local(args.num_proc, command, env=env,
                                  verbose=args.verbose)
    bad = [c for c in codes if c != 0]
    return bad[0] if bad else 0


if __name__ == "__main__":
    sys.exit(main())
