from .timing import CollectiveTimer, algbw_gbps, busbw_gbps

__all__ = ["CollectiveTimer", "algbw_gbps", "busbw_gbps"]
