#!/usr/bin/env bash
# Build/test driver (reference build.sh parity: clean/lib/tests/bench targets).
set -euo pipefail
cd "$(dirname "$0")"

target="${1:-lib}"

case "$target" in
  clean)
    rm -rf build raft_amd/_C*.so csrc/*.o
    ;;
  lib)
    python build_ext.py
    ;;
  tests)
    python -m pytest tests/ -q -m "not gpu"
    ;;
  gputests)
    python -m pytest tests/ -q -m gpu
    ;;
  bench)
    shift || true
    python bench.py "$@"
    ;;
  docs)
    PYTHONPATH=. python docs/gen_api.py
    ;;
  *)
    echo "usage: $0 {clean|lib|tests|gputests|bench|docs}" >&2
    exit 1
    ;;
esac
