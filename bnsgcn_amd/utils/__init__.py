from .timer import CommTimer, comm_timer

__all__ = ["CommTimer", "comm_timer"]
