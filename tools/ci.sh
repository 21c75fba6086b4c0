#!/bin/bash
# CI pipeline (the reference's Jenkinsfile check→test equivalent):
# build the extension + binaries, run the CPU test suite, then optional
# sanitizer passes (tools/sanitize.sh).
set -euo pipefail
cd "$(dirname "$0")/.."

echo "== build =="
python3 setup.py build_ext --inplace
make daemon

echo "== tests (CPU) =="
python3 -m pytest tests/ -x -q -m "not gpu" --timeout=240 --timeout-method=thread

echo "== bench smoke =="
timeout 300 python3 bench.py --gpus 1 --steps 5 --warmup 1 >/dev/null

echo "CI OK"
