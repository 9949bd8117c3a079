from ..shims.lpx_sklearn import ForestSoA

__all__ = ["ForestSoA"]
