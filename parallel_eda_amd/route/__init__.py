from .router import RouteResult, pathfinder_route, net_rr_terminals

__all__ = ["RouteResult", "pathfinder_route", "net_rr_terminals"]
