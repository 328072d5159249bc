"""Random generation: counter-based RNG, distributions, synthetic datasets.

Reference parity: raft/random/*  — PCG/Philox counter RNG
(rng_device.cuh), distribution transforms, make_blobs, make_regression, RMAT
graph generator, permute, multi-variable gaussian, sampling.
"""
from .rng import (
    RngState, uniform, uniform_int, normal, lognormal, logistic, exponential,
    rayleigh, laplace, gumbel, bernoulli, sample_with_replacement,
    sample_without_replacement,
)
from .make_blobs import make_blobs
from .make_regression import make_regression
from .rmat import rmat
from .permute import permute, permute_rows
from .mvg import multi_variable_gaussian

__all__ = [
    "RngState", "uniform", "uniform_int", "normal", "lognormal", "logistic",
    "exponential", "rayleigh", "laplace", "gumbel", "bernoulli",
    "sample_with_replacement", "sample_without_replacement",
    "make_blobs", "make_regression", "rmat", "permute", "permute_rows",
    "multi_variable_gaussian",
]
