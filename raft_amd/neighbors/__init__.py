from .fused_l2nn import fused_l2nn, fused_l2nn_argmin
from .brute_force import BruteForceIndex, brute_force_build, knn

__all__ = ["fused_l2nn", "fused_l2nn_argmin", "knn", "BruteForceIndex",
           "brute_force_build"]
