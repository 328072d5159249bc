"""Device bitset/bitmap over packed int32 words.

Reference parity: raft/core/bitset.hpp:33 (view) / :279 (owning) + bitmap.hpp:
set/test, flip, count via popcount, sparsity; used by masked-matmul and
bitmap->CSR conversion in sparse/.

Layout: little-endian bit order within 32-bit words, `n_words = ceil(n/32)`,
matching the reference so serialized masks are interchangeable.
"""
from __future__ import annotations

import torch


class Bitset:
    WORD_BITS = 32

    def __init__(self, n_bits: int, device=None, words: torch.Tensor | None = None,
                 default: bool = True):
        self.n_bits = int(n_bits)
        n_words = (self.n_bits + self.WORD_BITS - 1) // self.WORD_BITS
        if words is not None:
            assert words.numel() == n_words, "word buffer size mismatch"
            self.words = words.to(torch.int32)
        else:
            if device is None:
                device = "cpu"
            fill = -1 if default else 0
            self.words = torch.full((n_words,), fill, dtype=torch.int32, device=device)
            if default:
                self._mask_tail()

    def _mask_tail(self) -> None:
        rem = self.n_bits % self.WORD_BITS
        if rem and self.words.numel():
            tail_mask = (1 << rem) - 1
            # int32 arithmetic: keep only the valid low bits of the last word
            self.words[-1] = int(self.words[-1].item()) & tail_mask

    @property
    def device(self):
        return self.words.device

    # -- mutation ------------------------------------------------------------
    def set(self, idx: torch.Tensor, value: bool = True) -> None:
        """O(k) scatter set/clear. GPU: atomicOr/And kernel
        (csrc/histogram.hip); CPU: numpy bitwise-or.at — the round-1 version
        materialized the whole dense mask per call."""
        idx = idx.to(self.device, torch.int64)
        if self.words.is_cuda:
            from raft_amd._ext import require_ext
            self.words = self.words.contiguous()
            require_ext().bitset_set_(self.words, idx, bool(value))
            return
        import numpy as np
        w = self.words.numpy().view(np.uint32)
        word_idx = (idx // self.WORD_BITS).numpy()
        masks = (np.uint32(1) << (idx % self.WORD_BITS).numpy().astype(np.uint32))
        if value:
            np.bitwise_or.at(w, word_idx, masks)
        else:
            np.bitwise_and.at(w, word_idx, ~masks)

    def flip(self) -> None:
        self.words = ~self.words
        self._mask_tail()

    def reset(self, value: bool = False) -> None:
        self.words.fill_(-1 if value else 0)
        if value:
            self._mask_tail()

    # -- queries -------------------------------------------------------------
    def test(self, idx: torch.Tensor) -> torch.Tensor:
        idx = idx.to(self.device, torch.int64)
        if self.words.is_cuda:
            from raft_amd._ext import require_ext
            return require_ext().bitset_test(self.words.contiguous(), idx)
        word = idx // self.WORD_BITS
        bit = (idx % self.WORD_BITS).to(torch.int32)
        return ((self.words[word] >> bit) & 1).to(torch.bool)

    def to_dense(self) -> torch.Tensor:
        """Expand to a bool vector of length n_bits."""
        bits = torch.arange(self.WORD_BITS, device=self.device, dtype=torch.int32)
        expanded = ((self.words.unsqueeze(1) >> bits) & 1).to(torch.bool)
        return expanded.reshape(-1)[: self.n_bits]

    @classmethod
    def from_dense(cls, dense: torch.Tensor) -> "Bitset":
        dense = dense.to(torch.bool)
        n = dense.numel()
        n_words = (n + cls.WORD_BITS - 1) // cls.WORD_BITS
        pad = n_words * cls.WORD_BITS - n
        if pad:
            dense = torch.cat([dense, torch.zeros(pad, dtype=torch.bool, device=dense.device)])
        bits = dense.reshape(n_words, cls.WORD_BITS).to(torch.int64)
        weights = (torch.ones(cls.WORD_BITS, dtype=torch.int64, device=dense.device)
                   << torch.arange(cls.WORD_BITS, device=dense.device))
        words = (bits * weights).sum(dim=1)
        # wrap to int32 two's complement
        words = torch.where(words >= 2 ** 31, words - 2 ** 32, words).to(torch.int32)
        out = cls(n, device=dense.device, words=words)
        return out

    def count(self) -> int:
        """Population count (reference: bitset count via detail::popc).
        GPU: wave-reduced __popc kernel; CPU: byte popcount over the packed
        words (no dense expansion)."""
        if self.words.is_cuda:
            from raft_amd._ext import require_ext
            return int(require_ext().bitset_count(self.words.contiguous()).item())
        import numpy as np
        return int(np.unpackbits(self.words.numpy().view(np.uint8)).sum())

    def sparsity(self) -> float:
        return 1.0 - self.count() / max(self.n_bits, 1)

    def __len__(self) -> int:
        return self.n_bits
