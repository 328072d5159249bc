from .fused_l2nn import fused_l2nn, fused_l2nn_argmin
from .brute_force import knn

__all__ = ["fused_l2nn", "fused_l2nn_argmin", "knn"]
