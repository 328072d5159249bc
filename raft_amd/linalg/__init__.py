"""Dense linear algebra primitives (reference parity: raft/linalg/*).

mdspan-taking free functions become tensor-taking free functions. Hot reductions
and fused ops are hand-written HIP (wave64 logical-warp design) on GPU; plain
GEMMs go to rocBLAS/hipBLASLt through torch.matmul or the extension's
rocblas_gemm_ex wrapper (bf16-in/f32-out used by the fp32-emulation path).
"""
from .reduce import reduce, coalesced_reduction, strided_reduction, Apply
from .map import (
    map_op, map_offset, unary_op, binary_op, ternary_op,
    add, subtract, multiply, divide, power, sqrt, eltwise,
    map_then_reduce, map_reduce,
)
from .norm import norm, normalize, row_norm, col_norm, NormType
from .matrix_vector import matrix_vector_op, linewise_fused, linewise_op
from .gemm import gemm, gemv, dot, axpy, gemm_bf16_f32, gemm_fp32_emulated
from .reduce_by_key import reduce_rows_by_key, reduce_cols_by_key
from .decomp import eig, eigh, eig_jacobi, svd, qr, cholesky, cholesky_r1_update, lstsq
from .rsvd import rsvd
from .pca import pca_fit, pca_transform, pca_inverse_transform, tsvd_fit, tsvd_transform
from .misc import mean_squared_error, init_iota, init_eye, transpose

__all__ = [
    "reduce", "coalesced_reduction", "strided_reduction", "Apply",
    "map_op", "map_offset", "unary_op", "binary_op", "ternary_op",
    "add", "subtract", "multiply", "divide", "power", "sqrt", "eltwise",
    "map_then_reduce", "map_reduce",
    "norm", "normalize", "row_norm", "col_norm", "NormType",
    "matrix_vector_op", "linewise_op", "linewise_fused",
    "gemm", "gemv", "dot", "axpy", "gemm_bf16_f32", "gemm_fp32_emulated",
    "reduce_rows_by_key", "reduce_cols_by_key",
    "eig", "eigh", "eig_jacobi", "svd", "qr", "cholesky", "cholesky_r1_update", "lstsq",
    "rsvd", "pca_fit", "pca_transform", "pca_inverse_transform", "tsvd_fit", "tsvd_transform",
    "mean_squared_error", "init_iota", "init_eye", "transpose",
]
