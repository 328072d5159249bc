#!/usr/bin/env bash
# CI entry (reference ci/*.sh parity): hygiene -> build -> CPU suite.
# GPU lanes (pytest -m gpu, bench.py) run on MI355X boxes separately.
set -euo pipefail
cd "$(dirname "$0")/.."

echo "== include hygiene =="
python scripts/include_checker.py

echo "== native build (gfx950 cross-compile) =="
python build_ext.py

echo "== C++ library + out-of-tree consumer (mdspan API) =="
cmake -S . -B build/cpp -G Ninja > /dev/null
cmake --build build/cpp -j 16 > /dev/null
cmake --install build/cpp --prefix build/install > /dev/null
cmake -S tests/cpp/consumer -B build/consumer -G Ninja \
  -Draft_amd_DIR="$PWD/build/install/lib/cmake/raft_amd" > /dev/null
cmake --build build/consumer > /dev/null
echo "consumer built against installed package"

echo "== CPU test suite =="
python -m pytest tests/ -q -m "not gpu"

echo "== docs are current =="
PYTHONPATH=. python docs/gen_api.py > /dev/null
git diff --quiet -- docs/api || {
  echo "docs/api out of date: run docs/gen_api.py and commit"; exit 1; }

echo "CI OK"
