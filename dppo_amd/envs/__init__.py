from .synthetic import BatchedSyntheticEnv, make_env

__all__ = ["BatchedSyntheticEnv", "make_env"]
