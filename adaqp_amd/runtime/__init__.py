from .engine import GraphEngine
from .timer import Timer

__all__ = ['GraphEngine', 'Timer']
