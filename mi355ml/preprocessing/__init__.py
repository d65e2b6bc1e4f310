from .data import (MinMaxScaler, Normalizer, PolynomialFeatures,
                   QuantileTransformer, RobustScaler, StandardScaler)
from .label import LabelEncoder
from ._encoders import OneHotEncoder, OrdinalEncoder
from ._block_transformer import BlockTransformer
from ._dataframe import Categorizer, DummyEncoder

__all__ = [
    "StandardScaler", "MinMaxScaler", "RobustScaler", "QuantileTransformer",
    "Normalizer", "PolynomialFeatures", "LabelEncoder", "OneHotEncoder",
    "OrdinalEncoder", "BlockTransformer", "Categorizer", "DummyEncoder",
]
