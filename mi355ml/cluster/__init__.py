from .k_means import KMeans
from .spectral import SpectralClustering

__all__ = ["KMeans", "SpectralClustering"]
