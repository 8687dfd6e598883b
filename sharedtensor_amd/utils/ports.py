import socket


def free_port(host: str = "127.0.0.1") -> int:
    """Grab an ephemeral port that is currently free (best effort: the port
    is released again before returning, so races are possible; retry on
    bind failure)."""
    s = socket.socket()
    s.bind((host, 0))
    port = s.getsockname()[1]
    s.close()
    return port
