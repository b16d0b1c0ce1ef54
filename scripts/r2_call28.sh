#!/bin/bash
# Long endurance: ~11 min full-speed feedforward train with
# checkpoints + a kill/resume, finite-loss assertion at the end.
set -x
mkdir -p gpurun_out
O=$PWD/gpurun_out
SOAK_ARGS="--env synthetic-atari --num-actors 16 --envs-per-actor 16 \
  --batch-size 32 --unroll-length 20 --total-steps 2000000000 \
  --device cuda --xpid soak28 --savedir /tmp/soak28 \
  --checkpoint-interval-s 30"
python -m torched_impala_amd.cli $SOAK_ARGS > $O/r28_soak1.log 2>&1 &
PID=$!
sleep 420; kill -TERM $PID; wait $PID
tail -2 $O/r28_soak1.log
python -m torched_impala_amd.cli $SOAK_ARGS > $O/r28_soak2.log 2>&1 &
PID=$!
sleep 220; kill -TERM $PID; wait $PID
grep -h "resumed" $O/r28_soak2.log | head -1
tail -3 $O/r28_soak2.log
echo "nan count:"; grep -c "loss=nan" $O/r28_soak1.log $O/r28_soak2.log || true
echo "fps samples:"; grep -o "fps=[0-9]*" $O/r28_soak1.log | tail -3
