from . import dispatch, functional, reference

__all__ = ["dispatch", "functional", "reference"]
