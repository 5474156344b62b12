from .dist import (init_dist, spatial_partition, DistRouteLoop)

__all__ = ["init_dist", "spatial_partition", "DistRouteLoop"]
