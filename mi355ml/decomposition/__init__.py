from .pca import PCA
from .truncated_svd import TruncatedSVD
from .incremental_pca import IncrementalPCA

__all__ = ["PCA", "TruncatedSVD", "IncrementalPCA"]
