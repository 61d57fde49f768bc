#!/bin/bash
# Flake hunter (the ps-lite tests/repeat.sh analog): run the CPU suite
# N times, stop on first failure.
#   scripts/repeat_tests.sh 10            # 10 full runs
#   scripts/repeat_tests.sh 20 -k async   # 20 runs of a subset
set -e
N=${1:-5}; shift || true
cd "$(dirname "$0")/.."
for i in $(seq "$N"); do
  echo "== run $i/$N =="
  python -m pytest tests -q -m "not gpu" -x -p no:cacheprovider "$@"
done
echo "all $N runs green"
