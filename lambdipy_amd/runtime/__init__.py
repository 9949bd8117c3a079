from .coldstart import ColdStartResult, cold_start, prewarm_host

__all__ = ["ColdStartResult", "cold_start"]
