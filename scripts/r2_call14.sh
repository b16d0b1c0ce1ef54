#!/bin/bash
set -x
mkdir -p gpurun_out
O=$PWD/gpurun_out
timeout 400 python scripts/graph_drift_probe.py --steps 400 \
    > $O/r14_drift.log 2>&1
echo "DRIFT_RC=$?"
tail -24 $O/r14_drift.log
# lr=0 control: corruption without training dynamics?
timeout 260 python scripts/nan_probe.py --steps 4000 --learning-rate 0 \
    > $O/r14_lr0.log 2>&1
echo "LR0_RC=$?"
grep -h "first_nonfinite_step\|\"probe\"" $O/r14_lr0.log | tail -2
