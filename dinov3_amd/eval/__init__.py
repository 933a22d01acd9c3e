from .knn import evaluate_knn, extract_features
from .linear import evaluate_linear_probe

__all__ = ["extract_features", "evaluate_knn", "evaluate_linear_probe"]
