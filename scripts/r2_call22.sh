#!/bin/bash
# Post-fix full validation: suite, long probes (ff + lstm), headline
# re-measure, GPU kill/resume soak with finite-loss check.
set -x
mkdir -p gpurun_out
O=$PWD/gpurun_out

timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -3 \
    | tee $O/r22_gpu_tests.log

timeout 400 python scripts/nan_probe.py --steps 6000 \
    > $O/r22_probe_ff.log 2>&1
echo "FF_RC=$?"; grep -h '"probe"' $O/r22_probe_ff.log | tail -1
timeout 400 python scripts/nan_probe.py --steps 4000 --use-lstm \
    > $O/r22_probe_lstm.log 2>&1
echo "LSTM_RC=$?"; grep -h '"probe"' $O/r22_probe_lstm.log | tail -1

timeout 300 python scripts/learner_capacity.py --graphs --steps 100 \
    --warmup 20 > $O/r22_cap.json 2>/dev/null
timeout 600 python bench.py --steps 100 --warmup 20 \
    > $O/r22_bench.json 2>/dev/null
echo "BENCH_RC=$?"
timeout 600 python bench.py --steps 100 --warmup 20 --use-lstm \
    > $O/r22_bench_lstm.json 2>/dev/null

# kill/resume soak (fixed invocation), loss-finiteness asserted
SOAK_ARGS="--env synthetic-atari --num-actors 16 --envs-per-actor 16 \
  --batch-size 32 --unroll-length 20 --total-steps 2000000000 \
  --device cuda --xpid soak22 --savedir /tmp/soak \
  --checkpoint-interval-s 20"
python -m torched_impala_amd.cli $SOAK_ARGS > $O/r22_soak1.log 2>&1 &
PID=$!
sleep 90; kill -TERM $PID; wait $PID
echo "SOAK1 done"; tail -2 $O/r22_soak1.log
python -m torched_impala_amd.cli $SOAK_ARGS > $O/r22_soak2.log 2>&1 &
PID=$!
sleep 50; kill -TERM $PID; wait $PID
grep -h "resumed" $O/r22_soak2.log | head -1
tail -2 $O/r22_soak2.log
grep -c "loss=nan" $O/r22_soak1.log $O/r22_soak2.log || true

echo "=== JSON dump ==="
for f in $O/r22_cap.json $O/r22_bench.json $O/r22_bench_lstm.json; do
  echo "--- $f"; cat "$f" 2>/dev/null
done
