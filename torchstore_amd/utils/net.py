"""Small networking helpers (free-port picking, loopback resolution)."""

from __future__ import annotations

import socket


def pick_free_port(host: str = "127.0.0.1") -> int:
    """Bind port 0 to let the kernel choose, then release it.

    The tiny race between release and re-bind is acceptable for test/bootstrap
    use; long-lived servers bind port 0 directly and report the real port.
    """
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        s.bind((host, 0))
        return s.getsockname()[1]


def loopback() -> str:
    """Rendezvous address that resolves inside containers (hostname may not)."""
    return "127.0.0.1"
