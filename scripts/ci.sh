#!/usr/bin/env bash
# Full CPU-side check sequence (what the driver runs, plus the build).
set -euo pipefail
cd "$(dirname "$0")/.."
make -C native
python -c "from kubeshare_amd import ops; ops.build_extension()"
python -m pytest tests/ -q -m "not gpu"
python tools/simulator.py --jobs 200 --nodes 2 > /dev/null
echo "CI OK"
