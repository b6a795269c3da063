#!/usr/bin/env bash
# CI shard runner (the analog of the reference's scripts/tests.sh lanes).
# Usage: scripts/tests.sh [lane]
#   cpu       — the full CPU suite (default)
#   gpu       — GPU-marked tests (needs an MI355X)
#   bench     — benchmark harnesses
#   doctest   — doctest lane only
set -e
LANE="${1:-cpu}"
case "$LANE" in
  cpu)     exec python -m pytest tests/ -q -m "not gpu" ;;
  cpu-par)  exec python -m pytest tests/ -q -m "not gpu" -n auto ;;
  gpu)     exec python -m pytest tests/ -q -m gpu ;;
  bench)   exec python -m pytest benchmarks/ -q -s ;;
  doctest) exec python -m pytest tests/test_doctests.py -q ;;
  lint)    exec python -m pytest tests/test_formatting.py -q ;;
  sanitize) exec bash scripts/gpu_sanitize.sh ;;
  *) echo "unknown lane: $LANE" >&2; exit 2 ;;
esac
