#!/bin/bash
# Round-2 final validation bundle; each stage independent so one failure
# doesn't lose the rest.
set -x
mkdir -p gpurun_out
python scripts/measure_extra.py --sf 100 --reps 2 > gpurun_out/final_extra.log 2>&1
python scripts/measure_ds.py --sf 100 --reps 2 > gpurun_out/final_ds.log 2>&1
echo BUNDLE-DONE
