from .overlap import OverlappedReducer

__all__ = ["OverlappedReducer"]
