#!/bin/bash
# Reproduce every number in README.md / profiles/README.md on one MI355X.
# Usage: bash tools/run_all_benches.sh   (writes JSON lines to stdout)
set -e
cd "$(dirname "$0")/.."

echo "== build =="
python -c 'import __graft_entry__ as g; g.build()'

echo "== training benchmarks (driver contract: bench.py) =="
python bench.py --config cnn_dm --steps 200 --warmup 10
python bench.py --config cnn_dm --steps 12 --warmup 3 --batch 64
python bench.py --config lcsts --steps 50 --warmup 5
python bench.py --config longdoc --steps 6 --warmup 2

echo "== decode benchmarks =="
python scripts/bench_decode.py --n 24 --k 10

echo "== serving benchmark =="
python scripts/bench_serve.py --n 48 --concurrency 1 4 16

echo "== GPU test suite =="
python -m pytest tests/ -q -m gpu

echo "== kernel profile (rocprofv3; writes CSVs under /tmp/nats_prof) =="
cd /tmp && TMPDIR=/tmp rocprofv3 --kernel-trace --stats --output-format csv \
  -d /tmp/nats_prof -- python "$OLDPWD/bench.py" --steps 5 --warmup 2 || true
find /tmp/nats_prof -name "*kernel_stats*.csv" | head -1
