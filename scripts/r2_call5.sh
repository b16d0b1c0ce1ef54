#!/bin/bash
# Round-2 GPU call 5: MFMA-encoder-default validation — graph-mode
# capacity A/B, full suite, headline benches; box CPU inventory.
set -x
mkdir -p gpurun_out
R=$PWD
O=$PWD/gpurun_out
nproc | tee $O/r5_nproc.txt
grep "model name" /proc/cpuinfo | head -1 | tee -a $O/r5_nproc.txt

timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -6 \
    | tee $O/r5_gpu_tests.log

# graph-mode capacity: the product path. MFMA default vs fp32 A/B.
timeout 300 python scripts/learner_capacity.py --graphs --steps 100 \
    --warmup 20 > $O/r5_cap_graph_mfma.json 2> $O/r5_cap_graph_mfma.err
echo "CAPG_MFMA_RC=$?"
IMPALA_SHALLOW_MFMA=0 timeout 300 python scripts/learner_capacity.py \
    --graphs --steps 100 --warmup 20 \
    > $O/r5_cap_graph_fp32.json 2> $O/r5_cap_graph_fp32.err
echo "CAPG_FP32_RC=$?"
timeout 300 python scripts/learner_capacity.py --graphs --use-lstm \
    --steps 100 --warmup 20 \
    > $O/r5_cap_graph_lstm.json 2> $O/r5_cap_graph_lstm.err

timeout 600 python bench.py --steps 40 --warmup 12 \
    > $O/r5_bench.json 2> $O/r5_bench.err
echo "BENCH_RC=$?"
timeout 600 python bench.py --steps 40 --warmup 12 --use-lstm \
    > $O/r5_bench_lstm.json 2> $O/r5_bench_lstm.err
echo "BENCH_LSTM_RC=$?"

echo "=== JSON dump ==="
for f in $O/r5_cap_graph_mfma.json $O/r5_cap_graph_fp32.json \
         $O/r5_cap_graph_lstm.json $O/r5_bench.json $O/r5_bench_lstm.json; do
  echo "--- $f"; cat "$f" 2>/dev/null
done
tail -n 4 $O/r5_cap_graph_mfma.err 2>/dev/null
tail -n 3 $O/r5_bench.err
