import time
from typing import Callable


def wait_until(fn: Callable[[], bool], timeout: float = 30.0,
               interval: float = 0.05) -> bool:
    """Poll fn() until it returns True or timeout elapses."""
    t0 = time.time()
    while time.time() - t0 < timeout:
        if fn():
            return True
        time.sleep(interval)
    return False
