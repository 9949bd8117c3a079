from .coldstart import ColdStartResult, cold_start

__all__ = ["ColdStartResult", "cold_start"]
