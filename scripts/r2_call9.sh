#!/bin/bash
# Round-2 GPU call 9: bf16 LSTM weights + finalize/conv1_dw fixes.
set -x
mkdir -p gpurun_out
R=$PWD
O=$PWD/gpurun_out

./torched_impala_amd/csrc/deep_probe 2>&1 | tail -3 | tee $O/r9_probe.log
echo "PROBE_RC=${PIPESTATUS[0]}"

timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -4 \
    | tee $O/r9_gpu_tests.log

timeout 300 python scripts/learner_capacity.py --graphs --steps 100 \
    --warmup 20 > $O/r9_cap_graph.json 2> /dev/null
timeout 300 python scripts/learner_capacity.py --graphs --use-lstm \
    --steps 100 --warmup 20 > $O/r9_cap_graph_lstm.json 2> /dev/null
IMPALA_LSTM_BF16=0 timeout 300 python scripts/learner_capacity.py --graphs \
    --use-lstm --steps 100 --warmup 20 \
    > $O/r9_cap_graph_lstm_fp32w.json 2> /dev/null

timeout 600 python bench.py --steps 100 --warmup 20 \
    > $O/r9_bench.json 2> $O/r9_bench.err
echo "BENCH_RC=$?"
timeout 600 python bench.py --steps 100 --warmup 20 --use-lstm \
    > $O/r9_bench_lstm.json 2> /dev/null
echo "BENCH_LSTM_RC=$?"

cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats --output-format csv \
    -d /tmp/p9 -o g9 -- \
    python $R/scripts/learner_capacity.py --graphs --steps 60 --warmup 15 \
    > $O/r9_capg_prof.log 2>&1
find /tmp/p9 -name "*kernel_stats*" -exec cp {} $O/r9_graph_kernel_stats.csv \;
timeout 600 rocprofv3 --kernel-trace --stats --output-format csv \
    -d /tmp/p9l -o g9l -- \
    python $R/scripts/learner_capacity.py --graphs --use-lstm --steps 60 \
    --warmup 15 > $O/r9_capg_lstm_prof.log 2>&1
find /tmp/p9l -name "*kernel_stats*" -exec cp {} $O/r9_lstm_kernel_stats.csv \;
cd $R

echo "=== JSON dump ==="
for f in $O/r9_cap_graph.json $O/r9_cap_graph_lstm.json \
         $O/r9_cap_graph_lstm_fp32w.json $O/r9_bench.json \
         $O/r9_bench_lstm.json; do
  echo "--- $f"; cat "$f" 2>/dev/null
done
