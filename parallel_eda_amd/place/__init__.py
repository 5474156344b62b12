from .placer import Placement, anneal_place, analytic_delay_matrix

__all__ = ["Placement", "anneal_place", "analytic_delay_matrix"]
