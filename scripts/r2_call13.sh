#!/bin/bash
# Round-2 GPU call 13: NaN bisection matrix (r12 soak showed loss=nan).
set -x
mkdir -p gpurun_out
O=$PWD/gpurun_out

timeout 260 python scripts/nan_probe.py --steps 4000 \
    > $O/r13_default.log 2>&1
echo "=== default rc=$?"
grep -h "first_nonfinite_step\|\"probe\"" $O/r13_default.log | tail -2

IMPALA_SHALLOW_MFMA=0 timeout 260 python scripts/nan_probe.py --steps 4000 \
    > $O/r13_fp32enc.log 2>&1
echo "=== fp32enc rc=$?"
grep -h "first_nonfinite_step\|\"probe\"" $O/r13_fp32enc.log | tail -2

timeout 260 python scripts/nan_probe.py --steps 4000 --actor-jit frozen \
    > $O/r13_frozen.log 2>&1
echo "=== frozen rc=$?"
grep -h "first_nonfinite_step\|\"probe\"" $O/r13_frozen.log | tail -2

timeout 260 python scripts/nan_probe.py --steps 4000 --no-graphs \
    > $O/r13_nographs.log 2>&1
echo "=== nographs rc=$?"
grep -h "first_nonfinite_step\|\"probe\"" $O/r13_nographs.log | tail -2
