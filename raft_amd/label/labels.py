"""Label ops (reference: raft/label/*)."""
from __future__ import annotations

import torch


def get_unique_labels(labels: torch.Tensor) -> torch.Tensor:
    return torch.unique(labels)


def make_monotonic(labels: torch.Tensor, zero_based: bool = True) -> torch.Tensor:
    """Remap arbitrary label values to 0..n_unique-1 (sort-unique-map,
    classlabels.cuh:30)."""
    uniq, inv = torch.unique(labels, return_inverse=True)
    return inv if zero_based else inv + 1


def get_ovr_labels(labels: torch.Tensor, positive_class) -> torch.Tensor:
    """One-vs-rest binarization: +1 for the positive class, -1 otherwise."""
    return torch.where(labels == positive_class,
                       torch.ones_like(labels, dtype=torch.int64),
                       -torch.ones_like(labels, dtype=torch.int64))


def merge_labels(labels_a: torch.Tensor, labels_b: torch.Tensor,
                 mask: torch.Tensor | None = None) -> torch.Tensor:
    """Merge two labelings into connected equivalence classes.

    Reference parity: detail/merge_labels.cuh — iterated min-propagation over
    the implicit label-equivalence graph until fixpoint (propagate_label_kernel
    + reassign_label_kernel loop). Rows i, j get the same output label iff they
    are connected through shared labels in either labeling.
    """
    a = make_monotonic(labels_a)
    b = make_monotonic(labels_b)
    n = a.numel()
    na = int(a.max().item()) + 1 if n else 0
    nb = int(b.max().item()) + 1 if n else 0
    if mask is not None:
        m = mask.to(torch.bool)
    else:
        m = torch.ones(n, dtype=torch.bool, device=a.device)
    # node space: [rows 0..n) ∪ [a-classes n..n+na) ∪ [b-classes n+na..n+na+nb)
    total = n + na + nb
    parent = torch.arange(total, device=a.device)
    rows = torch.arange(n, device=a.device)

    # min-exchange fixpoint between rows and their two class nodes
    ca = a + n
    cb = b + n + na
    for _ in range(128):
        p = parent.clone()
        # rows adopt min of their two class parents (masked rows keep theirs)
        pr = torch.minimum(parent[rows], torch.minimum(parent[ca], parent[cb]))
        parent[rows] = torch.where(m, pr, parent[rows])
        # classes adopt min over member rows
        parent[n:n + na] = parent[n:n + na].scatter_reduce(
            0, a[m], parent[rows[m]], reduce="amin", include_self=True)
        parent[n + na:] = parent[n + na:].scatter_reduce(
            0, b[m], parent[rows[m]], reduce="amin", include_self=True)
        if torch.equal(p, parent):
            break
    out = parent[rows]
    return make_monotonic(out)
