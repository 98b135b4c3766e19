from . import imageutils, plotter

__all__ = ['plotter', 'imageutils']
