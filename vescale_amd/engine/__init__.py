from .pipe import PipeEngine

__all__ = ["PipeEngine"]
