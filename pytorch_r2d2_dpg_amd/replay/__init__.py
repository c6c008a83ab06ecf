from .sumtree_cpu import SumTree
from .sequence import SequenceCutter, Sequence
from .buffer import SequenceReplay

__all__ = ["SumTree", "SequenceCutter", "Sequence", "SequenceReplay"]
