"""Key-value pair for argmin/argmax reductions (reference: core/kvp.hpp).

The device-side equivalent lives in the HIP kernels as packed (f32 value,
i32 index) lanes (csrc/fused_l2nn.hip, csrc/reductions.hip row_argmin);
this Python type is the host-side surface for operators.argmin_op/argmax_op
and for APIs that return labelled extrema.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Any


@dataclass
class KeyValuePair:
    key: Any
    value: Any

    def __iter__(self):
        yield self.key
        yield self.value

    def __lt__(self, other: "KeyValuePair"):
        return (self.value, self.key) < (other.value, other.key)
