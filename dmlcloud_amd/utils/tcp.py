"""TCP rendezvous helpers for the MPI bootstrap path.

Capability parity with reference dmlcloud/util/tcp.py:5-27.
"""

import socket
import subprocess


def find_free_port() -> int:
    """Return a free TCP port on this machine."""
    with socket.socket() as s:
        s.bind(('', 0))
        return s.getsockname()[1]


def get_local_ips(use_hostname: bool = True):
    """Return the IP addresses of the local machine."""
    if use_hostname:
        proc = subprocess.run(['hostname', '-I'], capture_output=True, text=True)
        if proc.returncode == 0:
            return proc.stdout.strip().split(' ')
        raise RuntimeError(proc.stderr.strip())
    hostname = socket.gethostname()
    return socket.gethostbyname_ex(hostname)[2]
