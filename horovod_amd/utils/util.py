"""Small shared utilities (reference: horovod/common/util.py,
runner/util/hosts host hashing)."""
import hashlib
import socket


def split_list(xs, n):
    """Split xs into n contiguous chunks differing by at most one element."""
    k, m = divmod(len(xs), n)
    return [xs[i * k + min(i, m):(i + 1) * k + min(i + 1, m)]
            for i in range(n)]


def num_rank_is_power_2(num):
    return num != 0 and (num & (num - 1)) == 0


def host_hash(salt=None):
    """Stable per-host hash used to derive local ranks on heterogeneous
    launches."""
    h = socket.gethostname()
    if salt:
        h = f"{h}-{salt}"
    return int(hashlib.md5(h.encode()).hexdigest()[:8], 16)
