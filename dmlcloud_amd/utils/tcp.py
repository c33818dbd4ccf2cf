"""TCP rendezvous helpers for the MPI bootstrap path.

The MPI init ladder needs two things from this module: an unused port for
the rank-0 TCPStore, and an IP address of the rank-0 host that peers can
reach (capability parity with reference dmlcloud/util/tcp.py:5-27).
Address discovery here is pure-socket: the default-route trick first
(a connectionless UDP "connect" reveals which local interface the kernel
would route outbound traffic through — no packet is sent), then a
resolver lookup of the hostname, with `hostname -I` kept only as an
explicit opt-in.
"""

import socket
import subprocess
from typing import List

__all__ = ['find_free_port', 'get_local_ips']


def find_free_port(host: str = '') -> int:
    """Ask the kernel for an ephemeral TCP port and return it.

    The socket is closed before returning, so a race with other
    port-grabbers is possible but unlikely within a rendezvous window.
    """
    probe = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    try:
        probe.bind((host, 0))
        return probe.getsockname()[1]
    finally:
        probe.close()


def _default_route_ip() -> str:
    """IP of the interface the kernel routes outbound traffic through.

    UDP connect() only sets the destination — nothing is transmitted —
    so this works offline and never blocks.
    """
    probe = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
    try:
        probe.connect(('192.0.2.1', 9))  # TEST-NET-1: never routable
        return probe.getsockname()[0]
    finally:
        probe.close()


def _resolver_ips() -> List[str]:
    """All IPv4 addresses the resolver associates with this hostname."""
    infos = socket.getaddrinfo(socket.gethostname(), None, family=socket.AF_INET)
    seen = []
    for info in infos:
        addr = info[4][0]
        if addr not in seen:
            seen.append(addr)
    return seen


def get_local_ips(use_hostname: bool = False) -> List[str]:
    """IP addresses of this machine, most-routable first.

    Default path is pure-socket (default-route IP, then resolver
    entries, loopback filtered out unless it is all we have).
    ``use_hostname=True`` shells out to ``hostname -I`` instead — the
    flag exists for clusters whose resolver config lies about the
    fabric addresses.
    """
    if use_hostname:
        result = subprocess.run(['hostname', '-I'], capture_output=True, text=True)
        if result.returncode != 0:
            raise RuntimeError(f'hostname -I failed: {result.stderr.strip()}')
        return result.stdout.split()

    candidates: List[str] = []
    try:
        candidates.append(_default_route_ip())
    except OSError:
        pass
    try:
        for addr in _resolver_ips():
            if addr not in candidates:
                candidates.append(addr)
    except OSError:
        pass
    routable = [a for a in candidates if not a.startswith('127.')]
    if routable:
        return routable
    if candidates:
        return candidates
    return ['127.0.0.1']
