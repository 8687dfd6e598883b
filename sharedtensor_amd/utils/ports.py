import random
import socket

# Below the kernel's ephemeral range (Linux default 32768-60999): a listen
# port chosen here can never be stolen between probe and bind by some
# process's OUTGOING connection picking it as a source port — which is
# exactly what happens sporadically to "free" ports probed via bind(0).
_LOW, _HIGH = 20000, 30000


def free_port(host: str = "127.0.0.1", span: int = 1) -> int:
    """Return a port (for span > 1: the base of `span` consecutive ports,
    e.g. explicit tree topologies using base+rank) that is currently
    bindable on `host`, chosen outside the ephemeral range.  Best effort —
    the ports are released before returning; retry on bind failure."""
    for _ in range(500):
        base = random.randrange(_LOW, _HIGH - span)
        socks = []
        ok = True
        try:
            for i in range(span):
                s = socket.socket()
                s.bind((host, base + i))
                socks.append(s)
        except OSError:
            ok = False
        finally:
            for s in socks:
                s.close()
        if ok:
            return base
    raise RuntimeError("no free port span found")
