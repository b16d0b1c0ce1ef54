#!/bin/bash
# Round-2 GPU call 10: LSTM barrier cut + parity-test diagnostics.
set -x
mkdir -p gpurun_out
R=$PWD
O=$PWD/gpurun_out

timeout 300 python -m pytest tests/test_gpu_kernels.py -q -rA --tb=long \
    -k "parity_mfma or lstm_seq" 2>&1 | tail -60 \
    | tee $O/r10_parity_diag.log

timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -4 \
    | tee $O/r10_gpu_tests.log

timeout 300 python scripts/learner_capacity.py --graphs --steps 100 \
    --warmup 20 > $O/r10_cap_graph.json 2> /dev/null
timeout 300 python scripts/learner_capacity.py --graphs --use-lstm \
    --steps 100 --warmup 20 > $O/r10_cap_graph_lstm.json 2> /dev/null

timeout 600 python bench.py --steps 100 --warmup 20 \
    > $O/r10_bench.json 2> /dev/null
echo "BENCH_RC=$?"
timeout 600 python bench.py --steps 100 --warmup 20 --use-lstm \
    > $O/r10_bench_lstm.json 2> /dev/null

# actor supply after the shared-frame-bank fix
for A in 16 32 64; do
  timeout 300 python scripts/actor_probe.py --jit native-bf16 --actors $A \
      --seconds 8 > $O/r10_actor$A.json 2> /dev/null
done

echo "=== JSON dump ==="
for f in $O/r10_cap_graph.json $O/r10_cap_graph_lstm.json \
         $O/r10_bench.json $O/r10_bench_lstm.json $O/r10_actor16.json \
         $O/r10_actor32.json $O/r10_actor64.json; do
  echo "--- $f"; cat "$f" 2>/dev/null
done
