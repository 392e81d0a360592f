from .data import synthetic_batches, synthetic_eval_set, mnist_like_batches

__all__ = ["synthetic_batches", "synthetic_eval_set", "mnist_like_batches"]
