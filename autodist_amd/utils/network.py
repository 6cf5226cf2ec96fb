"""Local-address detection (reference autodist/utils/network.py:21-75, which
used netifaces; stdlib sockets here)."""
import socket
from typing import Set


def get_local_addresses() -> Set[str]:
    addrs = {"127.0.0.1", "localhost", "0.0.0.0"}
    try:
        hostname = socket.gethostname()
        addrs.add(hostname)
        for info in socket.getaddrinfo(hostname, None):
            addrs.add(info[4][0])
    except OSError:
        pass
    try:
        # address used for outbound traffic (no packets sent)
        with socket.socket(socket.AF_INET, socket.SOCK_DGRAM) as s:
            s.connect(("10.255.255.255", 1))
            addrs.add(s.getsockname()[0])
    except OSError:
        pass
    return addrs


def is_loopback_address(address: str) -> bool:
    return address in ("127.0.0.1", "localhost", "::1")


def is_local_address(address: str) -> bool:
    return address in get_local_addresses()
