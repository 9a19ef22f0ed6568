from baton_amd.models.mlp import LinearRegressionModel, TinyMLP, make_synthetic_regression

__all__ = ["LinearRegressionModel", "TinyMLP", "make_synthetic_regression"]
