#!/bin/bash
# CI examples runner (reference run_ci_examples.sh:25-46).
set -e

ROOT="$(cd "$(dirname "$0")" && pwd)"
export PYTHONPATH="$ROOT:${PYTHONPATH:-}"
pushd "$ROOT/examples" >/dev/null

for ex in simple.py simple_predict.py simple_objectstore.py \
          train_on_parquet.py readme_sklearn_api.py explainability.py \
          simple_partitioned.py simple_tune.py simple_dask.py \
          simple_modin.py simple_ray_dataset.py; do
    echo "=== examples/$ex ==="
    python "$ex"
done
python higgs.py --rows 100000 --actors 2

popd >/dev/null
echo "CI EXAMPLES PASSED"
