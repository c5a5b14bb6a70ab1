from .comm import Comm, FlatBuffers

__all__ = ["Comm", "FlatBuffers"]
