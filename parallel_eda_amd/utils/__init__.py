from .stats import StatsWriter, routing_stats
from .checkpoint import save_router_state, load_router_state

__all__ = ["StatsWriter", "routing_stats", "save_router_state",
           "load_router_state"]
