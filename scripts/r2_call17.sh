#!/bin/bash
# Episode-wave discriminator + non-finite state capture.
set -x
mkdir -p gpurun_out
O=$PWD/gpurun_out

# short episodes: waves at learner step ~40k+3 -> NaN should move EARLY
timeout 200 python scripts/nan_probe.py --steps 2000 \
    --episode-len 100 > $O/r17_ep100.log 2>&1
echo "EP100_RC=$?"
grep -h "first_nonfinite_step" $O/r17_ep100.log | tail -1
# long episodes: first wave at ~3970 -> expect clean until then
timeout 260 python scripts/nan_probe.py --steps 3500 \
    --episode-len 10000 > $O/r17_ep10k.log 2>&1
echo "EP10K_RC=$?"
grep -h "first_nonfinite_step\|\"probe\"" $O/r17_ep10k.log | tail -1
# capture the poisoned state at default settings
IMPALA_DUMP_NONFINITE=$O/r17_dump.pt timeout 260 python \
    scripts/nan_probe.py --steps 1600 > $O/r17_dump_run.log 2>&1
echo "DUMP_RC=$?"
ls -la $O/r17_dump.pt 2>/dev/null
grep -h "first_nonfinite_step" $O/r17_dump_run.log | tail -1
