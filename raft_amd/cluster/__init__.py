from .kmeans import (KMeans, KMeansParams, kmeans_fit, kmeans_predict,
                     kmeans_transform, kmeans_iterate)

__all__ = ["KMeans", "KMeansParams", "kmeans_fit", "kmeans_predict", "kmeans_transform", "kmeans_iterate"]
