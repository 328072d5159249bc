"""Label utilities (reference: raft/label/classlabels.cuh:30-104
make_monotonic/getUniquelabels/getOvrlabels; detail/merge_labels.cuh iterated
min-propagation over the label-equivalence graph)."""
from .labels import make_monotonic, get_unique_labels, get_ovr_labels, merge_labels

__all__ = ["make_monotonic", "get_unique_labels", "get_ovr_labels", "merge_labels"]
