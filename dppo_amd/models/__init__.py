from .mlp import Model, PolicyValueMLP, normc_init_

__all__ = ["Model", "PolicyValueMLP", "normc_init_"]
