"""Pick an OS-assigned free TCP port for gloo rendezvous — fixed ports
collide with lingering sockets when suites run back to back."""

import socket


def free_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]
