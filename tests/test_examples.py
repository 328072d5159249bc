"""Examples must stay runnable (smoke, tiny shapes, CPU path)."""
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.parametrize("script,args", [
    ("kmeans_example.py", ["--rows", "3000", "--k", "8", "--dim", "16"]),
    ("knn_example.py", ["--index-rows", "5000", "--queries", "100", "--k", "5"]),
    ("eigsh_example.py", ["--n", "1500", "--k", "4"]),
    ("pairwise_topk_example.py", []),
])
def test_example_runs(script, args):
    r = subprocess.run([sys.executable, os.path.join(ROOT, "examples", script), *args],
                       capture_output=True, timeout=600)
    assert r.returncode == 0, r.stderr.decode()[-2000:]
    assert b"device=" in r.stdout
