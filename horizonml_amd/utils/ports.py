"""Rendezvous-port helper (reference ``data_parallel_train.py:22-25``)."""
from __future__ import annotations

import socket


def find_free_port() -> int:
    """Bind an ephemeral socket on 127.0.0.1 and return its port."""
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]
