from .stack import MACEStack
from . import o3, blocks, symmetric_contraction
