from .mlp import MLP, MlpFunction, mlp_function

__all__ = ["MLP", "MlpFunction", "mlp_function"]
