from . import dist

__all__ = ['dist']
