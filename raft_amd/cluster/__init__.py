from .kmeans import (KMeans, KMeansParams, kmeans_fit, kmeans_predict,
                     kmeans_transform, kmeans_iterate, kmeans_iter_state, kmeans_balanced_fit)

__all__ = ["KMeans", "KMeansParams", "kmeans_fit", "kmeans_predict", "kmeans_transform", "kmeans_iterate", "kmeans_iter_state", "kmeans_balanced_fit"]
