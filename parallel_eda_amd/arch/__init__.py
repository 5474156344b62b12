from .archdef import ArchDef, BUILTIN_ARCHES, get_arch

__all__ = ["ArchDef", "BUILTIN_ARCHES", "get_arch"]
