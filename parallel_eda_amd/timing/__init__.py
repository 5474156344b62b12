from .sta import STA

__all__ = ["STA"]
