#!/bin/bash
# Round-2 GPU call 8: grid-stride shallow dW validation + LSTM-mode
# rocprof breakdown + longer steady-state bench.
set -x
mkdir -p gpurun_out
R=$PWD
O=$PWD/gpurun_out

./torched_impala_amd/csrc/deep_probe 2>&1 | tail -12 | tee $O/r8_probe.log
echo "PROBE_RC=${PIPESTATUS[0]}"
./torched_impala_amd/csrc/deep_probe bench 2>&1 | tail -4 \
    | tee $O/r8_probe_bench.log

timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -4 \
    | tee $O/r8_gpu_tests.log

timeout 300 python scripts/learner_capacity.py --graphs --steps 100 \
    --warmup 20 > $O/r8_cap_graph.json 2> /dev/null
timeout 300 python scripts/learner_capacity.py --graphs --use-lstm \
    --steps 100 --warmup 20 > $O/r8_cap_graph_lstm.json 2> /dev/null

timeout 600 python bench.py --steps 100 --warmup 20 \
    > $O/r8_bench100.json 2> $O/r8_bench100.err
echo "BENCH100_RC=$?"
timeout 600 python bench.py --steps 40 --warmup 12 --use-lstm \
    > $O/r8_bench_lstm.json 2> /dev/null

cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats --output-format csv \
    -d /tmp/p8 -o g8 -- \
    python $R/scripts/learner_capacity.py --graphs --steps 60 --warmup 15 \
    > $O/r8_capg_prof.log 2>&1
find /tmp/p8 -name "*kernel_stats*" -exec cp {} $O/r8_graph_kernel_stats.csv \;
timeout 600 rocprofv3 --kernel-trace --stats --output-format csv \
    -d /tmp/p8l -o g8l -- \
    python $R/scripts/learner_capacity.py --graphs --use-lstm --steps 60 \
    --warmup 15 > $O/r8_capg_lstm_prof.log 2>&1
find /tmp/p8l -name "*kernel_stats*" -exec cp {} $O/r8_lstm_kernel_stats.csv \;
cd $R

echo "=== JSON dump ==="
for f in $O/r8_cap_graph.json $O/r8_cap_graph_lstm.json \
         $O/r8_bench100.json $O/r8_bench_lstm.json; do
  echo "--- $f"; cat "$f" 2>/dev/null
done
