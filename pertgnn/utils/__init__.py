from .logging import JsonlLogger

__all__ = ["JsonlLogger"]
