"""Pick a genuinely free TCP port for gloo rendezvous (random ports can
collide with TIME_WAIT sockets of earlier tests in the same run)."""

import socket


def free_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]
