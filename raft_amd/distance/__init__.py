from .pairwise import pairwise_distance, DistanceType

__all__ = ["pairwise_distance", "DistanceType"]
