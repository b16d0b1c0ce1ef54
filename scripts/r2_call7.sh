#!/bin/bash
# Round-2 GPU call 7: conv1 rewrite + fused bwd_finalize validation.
set -x
mkdir -p gpurun_out
R=$PWD
O=$PWD/gpurun_out

./torched_impala_amd/csrc/deep_probe 2>&1 | tail -14 | tee $O/r7_probe.log
echo "PROBE_RC=${PIPESTATUS[0]}"
./torched_impala_amd/csrc/deep_probe bench 2>&1 | tail -10 \
    | tee $O/r7_probe_bench.log

timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -5 \
    | tee $O/r7_gpu_tests.log

timeout 300 python scripts/learner_capacity.py --graphs --steps 100 \
    --warmup 20 > $O/r7_cap_graph.json 2> $O/r7_cap_graph.err
echo "CAPG_RC=$?"
timeout 300 python scripts/learner_capacity.py --graphs --use-lstm \
    --steps 100 --warmup 20 > $O/r7_cap_graph_lstm.json 2> /dev/null

timeout 600 python bench.py --steps 40 --warmup 12 \
    > $O/r7_bench.json 2> $O/r7_bench.err
echo "BENCH_RC=$?"
timeout 600 python bench.py --steps 40 --warmup 12 --use-lstm \
    > $O/r7_bench_lstm.json 2> /dev/null

cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats --output-format csv \
    -d /tmp/p_g7 -o g7 -- \
    python $R/scripts/learner_capacity.py --graphs --steps 60 --warmup 15 \
    > $O/r7_capg_prof.log 2>&1
find /tmp/p_g7 -name "*kernel_stats*" -exec cp {} $O/r7_graph_kernel_stats.csv \;
cd $R

echo "=== JSON dump ==="
for f in $O/r7_cap_graph.json $O/r7_cap_graph_lstm.json $O/r7_bench.json \
         $O/r7_bench_lstm.json; do
  echo "--- $f"; cat "$f" 2>/dev/null
done
tail -n 4 $O/r7_cap_graph.err $O/r7_bench.err 2>/dev/null
