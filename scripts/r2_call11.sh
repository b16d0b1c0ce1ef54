#!/bin/bash
# Round-2 GPU call 11: finalize-grid fix validation + clean re-measure.
set -x
mkdir -p gpurun_out
R=$PWD
O=$PWD/gpurun_out

timeout 300 python -m pytest tests/test_gpu_kernels.py -q \
    -k "parity_mfma" 2>&1 | tail -6 | tee $O/r11_parity.log

timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -4 \
    | tee $O/r11_gpu_tests.log

timeout 300 python scripts/learner_capacity.py --graphs --steps 100 \
    --warmup 20 > $O/r11_cap_graph.json 2> /dev/null
timeout 300 python scripts/learner_capacity.py --graphs --use-lstm \
    --steps 100 --warmup 20 > $O/r11_cap_graph_lstm.json 2> /dev/null

timeout 600 python bench.py --steps 100 --warmup 20 \
    > $O/r11_bench.json 2> /dev/null
echo "BENCH_RC=$?"
timeout 600 python bench.py --steps 100 --warmup 20 --use-lstm \
    > $O/r11_bench_lstm.json 2> /dev/null

# short GPU endurance: 1000 learner steps of real training w/ actors
timeout 600 python -m torched_impala_amd.cli train --env synthetic-atari \
    --num-actors 16 --envs-per-actor 16 --batch-size 32 --unroll-length 20 \
    --total-steps 640000 --device cuda --xpid soak_r11 \
    --savedir /tmp/soak 2>&1 | tail -5 | tee $O/r11_soak.log
echo "SOAK_RC=$?"

echo "=== JSON dump ==="
for f in $O/r11_cap_graph.json $O/r11_cap_graph_lstm.json \
         $O/r11_bench.json $O/r11_bench_lstm.json; do
  echo "--- $f"; cat "$f" 2>/dev/null
done
