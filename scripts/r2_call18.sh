#!/bin/bash
# Hunt the poisoned step + validate the grad-skip guard end-to-end.
set -x
mkdir -p gpurun_out
O=$PWD/gpurun_out

# 1. hunt mode: no per-step sync; dump the poisoned batch window.
#    With the skip guard the run should now SURVIVE even if grads spike.
timeout 500 python scripts/nan_probe.py --steps 6000 \
    --hunt $O/r18_hunt.pt > $O/r18_hunt.log 2>&1
echo "HUNT_RC=$?"
grep -h "hunt_first_bad\|\"probe\"" $O/r18_hunt.log | tail -2
ls -la $O/r18_hunt.pt 2>/dev/null | awk '{print $5, $9}'

# 2. plain probe (per-step sync): grad_skip_count visibility
timeout 400 python scripts/nan_probe.py --steps 5000 \
    > $O/r18_plain.log 2>&1
echo "PLAIN_RC=$?"
grep -h "first_nonfinite_step\|\"probe\"" $O/r18_plain.log | tail -1

# 3. if the hunt dumped: replay the batch deterministically
if [ -s $O/r18_hunt.pt ]; then
  timeout 300 python scripts/dump_replay.py $O/r18_hunt.pt \
      > $O/r18_replay.log 2>&1
  echo "REPLAY_RC=$?"
  tail -20 $O/r18_replay.log
fi
