"""NIC discovery for multi-host launches.

Reference: horovod/runner/driver/driver_service.py:30-200 — before
launching, the driver probes which network interfaces are routable from
every host and passes the common set to NCCL/Gloo.  The reference runs a
driver/task RPC service pair; this launcher is ssh-based, so the probe
rides ssh: each remote host reports its interfaces + addresses, and the
launcher intersects them (same outcome, no extra service processes).
"""
import json
import socket
import struct
import subprocess

EXCLUDE_PREFIXES = ("lo", "docker", "veth", "br-", "virbr", "tun", "tap")


def local_interfaces():
    """{ifname: ipv4} for this machine's plausible data-plane NICs."""
    try:
        import fcntl
    except ImportError:  # non-Linux: best effort via hostname
        return {"default": socket.gethostbyname(socket.gethostname())}
    out = {}
    s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
    try:
        for _, name in socket.if_nameindex():
            if name.startswith(EXCLUDE_PREFIXES):
                continue
            try:
                packed = struct.pack("256s", name[:15].encode())
                addr = socket.inet_ntoa(fcntl.ioctl(
                    s.fileno(), 0x8915, packed)[20:24])  # SIOCGIFADDR
            except OSError:
                continue
            out[name] = addr
    finally:
        s.close()
    return out


_PROBE_SNIPPET = (
    "import json,socket,struct,fcntl\n"
    "o={}\n"
    "s=socket.socket(socket.AF_INET,socket.SOCK_DGRAM)\n"
    "for _,n in socket.if_nameindex():\n"
    "  if n.startswith(%r): continue\n"
    "  try:\n"
    "    a=socket.inet_ntoa(fcntl.ioctl(s.fileno(),0x8915,"
    "struct.pack('256s',n[:15].encode()))[20:24])\n"
    "  except OSError: continue\n"
    "  o[n]=a\n"
    "print(json.dumps(o))\n" % (EXCLUDE_PREFIXES,))


def remote_interfaces(host, python="python3", timeout=20):
    """Probe a remote host's NICs over ssh (the reference's task-service
    register_task_to_task_addresses analogue)."""
    import base64
    b64 = base64.b64encode(_PROBE_SNIPPET.encode()).decode()
    cmd = ["ssh", "-o", "StrictHostKeyChecking=no",
           "-o", "ConnectTimeout=10", host, python, "-c",
           f"\"import base64;exec(base64.b64decode('{b64}'))\""]
    try:
        out = subprocess.run(cmd, capture_output=True, text=True,
                             timeout=timeout)
        if out.returncode != 0:
            return {}
        return json.loads(out.stdout.strip().splitlines()[-1])
    except Exception:
        return {}


def find_common_interfaces(hosts, python="python3", verbose=False):
    """Interface names present (with an IPv4 address) on EVERY host
    (reference: driver_service.py _run_probe common-intersection).  Local
    host specs (localhost/127.0.0.1) use the local probe."""
    common = None
    for host in hosts:
        if host in ("localhost", "127.0.0.1"):
            ifaces = local_interfaces()
        else:
            ifaces = remote_interfaces(host, python=python)
        names = set(ifaces)
        if verbose:
            print(f"[hvdrun] {host}: interfaces {sorted(names)}")
        if not names:
            continue
        common = names if common is None else (common & names)
    return sorted(common) if common else []


def resolve_nics(args_interface, hosts, verbose=False):
    """The launcher's NIC decision: an explicit --network-interface wins;
    otherwise multi-host launches probe for the common set and single-host
    launches need none (loopback)."""
    if args_interface:
        return args_interface
    remote = [h for h in hosts if h not in ("localhost", "127.0.0.1")]
    if not remote:
        return None
    common = find_common_interfaces(hosts, verbose=verbose)
    return ",".join(common) if common else None
