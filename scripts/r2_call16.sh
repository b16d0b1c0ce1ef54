#!/bin/bash
# Validate the capture-all-before-replay fix.
set -x
mkdir -p gpurun_out
O=$PWD/gpurun_out

# 1. old behavior, 2-buffer train pattern: expect corruption
IMPALA_GRAPH_LAZY_CAPTURE=1 timeout 420 python scripts/graph_drift_probe.py \
    --stress --buffers 2 --steps 5000 > $O/r16_lazy2.log 2>&1
echo "LAZY2_RC=$? (nonzero = corruption reproduced)"
tail -4 $O/r16_lazy2.log
# 2. fixed behavior, same pattern: expect clean
timeout 420 python scripts/graph_drift_probe.py --stress --buffers 2 \
    --steps 5000 > $O/r16_fixed2.log 2>&1
echo "FIXED2_RC=$?"
tail -3 $O/r16_fixed2.log
# 3. the decisive end-to-end check: default train path, 4000 steps
timeout 260 python scripts/nan_probe.py --steps 4000 \
    > $O/r16_nan_default.log 2>&1
echo "NAN_DEFAULT_RC=$?"
grep -h "first_nonfinite_step\|\"probe\"" $O/r16_nan_default.log | tail -1
# 4. suite
timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -3 \
    | tee $O/r16_gpu_tests.log
