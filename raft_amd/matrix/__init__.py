"""Matrix ops: k-selection, gather/scatter, structural ops, sort, argmin/max.

Reference parity: raft/matrix/*  — most notably the select_k
engine (radix + warpsort) that the reference's ANN stack is built on.
"""
from .select_k import select_k, SelectAlgo
from .gather import gather, gather_if, scatter, gather_inplace, scatter_inplace
from .argminmax import argmax, argmin
from .ops import (
    slice_matrix, get_diagonal, set_diagonal, upper_triangular, lower_triangular,
    row_reverse, col_reverse, shift_rows, eye, power, ratio, reciprocal,
    sqrt as matrix_sqrt, sign_flip, threshold, linewise,
)
from .sort import col_wise_sort
from .sample_rows import sample_rows
from .norm import l2_norm
from .print import print_matrix

__all__ = [
    "select_k", "SelectAlgo", "gather", "gather_if", "scatter",
    "gather_inplace", "scatter_inplace",
    "argmax", "argmin", "slice_matrix", "get_diagonal", "set_diagonal",
    "upper_triangular", "lower_triangular", "row_reverse", "col_reverse",
    "shift_rows", "eye", "power", "ratio", "reciprocal", "matrix_sqrt",
    "sign_flip", "threshold", "linewise", "col_wise_sort", "sample_rows", "l2_norm", "print_matrix",
]
