#!/bin/bash
# CI test runner (reference run_ci_tests.sh:23-38 protocol:
# per-file pytest -x + a benchmark smoke run).
set -e

ROOT="$(cd "$(dirname "$0")" && pwd)"
export PYTHONPATH="$ROOT:${PYTHONPATH:-}"
pushd "$ROOT" >/dev/null

for f in tests/test_*.py; do
    echo "=== $f ==="
    python -m pytest -x -q -m "not gpu" "$f" || {
        rc=$?
        # 5 = file contains only gpu-marked tests (all deselected)
        [ "$rc" -eq 5 ] || exit "$rc"
    }
done

echo "=== benchmark smoke ==="
python benchmarks/benchmark_cpu_gpu.py 2 10 20 --smoke-test
python benchmarks/benchmark_ft.py --smoke-test

popd >/dev/null
echo "CI TESTS PASSED"
