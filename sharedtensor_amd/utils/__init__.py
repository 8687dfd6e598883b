from .ports import free_port
from .timing import wait_until

__all__ = ["free_port", "wait_until"]
