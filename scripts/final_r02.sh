#!/bin/bash
# Round-2 final validation bundle; each stage independent.
set -x
mkdir -p gpurun_out
python -m pytest tests -m gpu -q 2>&1 | tail -3 > gpurun_out/final_pytest.txt
python -c "import __graft_entry__ as g; g.smoke(); print('SMOKE OK')" \
  2>&1 | tail -2 > gpurun_out/final_smoke.txt
python bench.py --steps 10 --warmup 3 --query all \
  > gpurun_out/final_bench.json 2> gpurun_out/final_bench.err
python scripts/measure_extra.py --sf 100 --reps 2 > gpurun_out/final_extra.log 2>&1
python scripts/measure_ds.py --sf 100 --reps 2 > gpurun_out/final_ds.log 2>&1
echo BUNDLE-DONE
