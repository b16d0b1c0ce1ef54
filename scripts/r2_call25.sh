#!/bin/bash
# Final round-2 evidence bundle.
set -x
mkdir -p gpurun_out
R=$PWD
O=$PWD/gpurun_out

timeout 600 python bench.py --steps 200 --warmup 30 \
    > $O/r25_bench200.json 2>/dev/null
echo BENCH_RC=$?
timeout 600 python bench.py --steps 200 --warmup 30 --use-lstm \
    > $O/r25_bench200_lstm.json 2>/dev/null

cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats --output-format csv \
    -d /tmp/p25 -o g25 -- \
    python $R/scripts/learner_capacity.py --graphs --steps 60 --warmup 15 \
    > $O/r25_capg_prof.log 2>&1
find /tmp/p25 -name "*kernel_stats*" -exec cp {} $O/r25_final_kernel_stats.csv \;
timeout 600 rocprofv3 --kernel-trace --stats --output-format csv \
    -d /tmp/p25l -o g25l -- \
    python $R/scripts/learner_capacity.py --graphs --use-lstm --steps 60 \
    --warmup 15 > $O/r25_capg_lstm_prof.log 2>&1
find /tmp/p25l -name "*kernel_stats*" -exec cp {} $O/r25_final_lstm_kernel_stats.csv \;
cd $R

# 4-minute LSTM kill/resume endurance with finite-loss check
SOAK_ARGS="--env synthetic-atari --use-lstm --num-actors 16 \
  --envs-per-actor 16 --batch-size 32 --unroll-length 20 \
  --total-steps 2000000000 --device cuda --xpid soak25 \
  --savedir /tmp/soak25 --checkpoint-interval-s 20"
python -m torched_impala_amd.cli $SOAK_ARGS > $O/r25_soak1.log 2>&1 &
PID=$!
sleep 150; kill -TERM $PID; wait $PID
tail -2 $O/r25_soak1.log
python -m torched_impala_amd.cli $SOAK_ARGS > $O/r25_soak2.log 2>&1 &
PID=$!
sleep 90; kill -TERM $PID; wait $PID
grep -h "resumed" $O/r25_soak2.log | head -1
tail -2 $O/r25_soak2.log
echo "nan lines:"; grep -c "loss=nan" $O/r25_soak1.log $O/r25_soak2.log || true
for f in $O/r25_bench200.json $O/r25_bench200_lstm.json; do echo "--- $f"; cat $f; done
