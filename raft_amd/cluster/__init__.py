from .kmeans import KMeans, KMeansParams, kmeans_fit, kmeans_predict, kmeans_transform

__all__ = ["KMeans", "KMeansParams", "kmeans_fit", "kmeans_predict", "kmeans_transform"]
