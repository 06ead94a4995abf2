#!/bin/bash
# Full CPU-side verification lane: build, tests, ASan, bench contract.
set -euo pipefail
cd "$(dirname "$0")/.."
echo "== build (gfx950 cross-compile) =="
PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace >/dev/null
python -c "import __graft_entry__"
echo "== CPU test tier =="
python -m pytest tests -q -m "not gpu"
echo "== ASan lane (C++ log collector) =="
bash scripts/asan_check.sh
echo "== examples =="
for ex in examples/*.py; do
  echo "-- $ex"
  python "$ex" >/dev/null
done
echo "== bench contract (CPU tiny) =="
python bench.py --steps 2 --warmup 1 | python -c "
import json, sys
d = json.loads(sys.stdin.read().strip().splitlines()[-1])
for key in ('metric','value','unit','n_gpus','steps','warmup',
            'ms_per_step','higher_is_better','scaling','dtype','config'):
    assert key in d, key
print('bench JSON contract ok')"
echo "ALL CHECKS PASSED"
