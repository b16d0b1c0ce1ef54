#!/bin/bash
# Round-2 GPU call 12: deep-config attribution + GPU kill/resume soak.
set -x
mkdir -p gpurun_out
R=$PWD
O=$PWD/gpurun_out

# ---- GPU kill/resume endurance (checkpointing every 20 s) ----
SOAK_ARGS="--env synthetic-atari --num-actors 16 --envs-per-actor 16 \
  --batch-size 32 --unroll-length 20 --total-steps 2000000000 \
  --device cuda --xpid soak12 --savedir /tmp/soak \
  --checkpoint-interval-s 20"
python -m torched_impala_amd.cli $SOAK_ARGS \
    > $O/r12_soak1.log 2>&1 &
PID=$!
sleep 100
kill -TERM $PID
wait $PID
echo "SOAK1_RC=$? (killed by SIGTERM = expected)"
tail -3 $O/r12_soak1.log
python -m torched_impala_amd.cli $SOAK_ARGS \
    > $O/r12_soak2.log 2>&1 &
PID=$!
sleep 60
kill -TERM $PID
wait $PID
echo "SOAK2_RC=$?"
grep -h "resumed" $O/r12_soak2.log | head -2
tail -3 $O/r12_soak2.log

# ---- deep big-config rocprof attribution (graphs) ----
cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats --output-format csv \
    -d /tmp/pd -o d12 -- \
    python $R/scripts/learner_capacity.py --model deep --use-lstm \
    --batch-size 512 --unroll-length 100 --steps 4 --warmup 1 --graphs \
    > $O/r12_deep_prof.log 2>&1
echo "DEEP_PROF_RC=$?"
find /tmp/pd -name "*kernel_stats*" -exec cp {} $O/r12_deep_kernel_stats.csv \;
cd $R
grep -h "ms_per_step" $O/r12_deep_prof.log
du -sh $O
