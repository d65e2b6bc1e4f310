from .data import StandardScaler, MinMaxScaler

__all__ = ["StandardScaler", "MinMaxScaler"]
