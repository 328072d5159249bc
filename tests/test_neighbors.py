import pytest
import torch

from raft_amd.neighbors import knn
from raft_amd.distance import DistanceType


class TestKnn:
    @pytest.mark.parametrize("qc,ic", [(1000, 1000000), (7, 13)])
    def test_matches_cdist(self, qc, ic):
        torch.manual_seed(0)
        x = torch.randn(100, 8)
        q = torch.randn(25, 8)
        d, i = knn(x, q, k=5, query_chunk=qc, index_chunk=ic)
        ref = torch.cdist(q.double(), x.double()) ** 2
        rd, ri = torch.topk(ref, 5, dim=1, largest=False)
        torch.testing.assert_close(d.double(), rd, rtol=1e-4, atol=1e-4)
        # allow tie permutations: check distances via gather
        torch.testing.assert_close(ref.gather(1, i), rd, rtol=1e-4, atol=1e-4)

    def test_chunked_merge_correct(self):
        """chunking across the index dim must not change results."""
        torch.manual_seed(1)
        x = torch.randn(200, 4)
        q = torch.randn(10, 4)
        d1, i1 = knn(x, q, k=8, index_chunk=37)
        d2, i2 = knn(x, q, k=8, index_chunk=1000000)
        torch.testing.assert_close(d1, d2, rtol=1e-4, atol=1e-5)
        assert torch.equal(i1, i2)
