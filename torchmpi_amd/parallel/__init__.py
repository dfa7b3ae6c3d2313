from .communicator import Communicator, ring_permutations
from .hierarchical import hierarchical_allreduce

__all__ = ["Communicator", "ring_permutations", "hierarchical_allreduce"]
