#!/bin/bash
# Round-2 closing measurements (one box, full matrix).
set -x
mkdir -p gpurun_out
O=$PWD/gpurun_out
timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -3 | tee $O/final_tests.log
timeout 600 python bench.py --steps 200 --warmup 30 > $O/final_bench.json 2>/dev/null
timeout 600 python bench.py --steps 200 --warmup 30 --use-lstm > $O/final_bench_lstm.json 2>/dev/null
timeout 300 python scripts/learner_capacity.py --graphs --steps 150 --warmup 30 > $O/final_cap.json 2>/dev/null
timeout 300 python scripts/learner_capacity.py --graphs --use-lstm --steps 150 --warmup 30 > $O/final_cap_lstm.json 2>/dev/null
timeout 600 python scripts/learner_capacity.py --model deep --use-lstm \
    --batch-size 512 --unroll-length 100 --steps 8 --warmup 3 --graphs \
    > $O/final_cap_deep.json 2>/dev/null
timeout 260 python scripts/nan_probe.py --steps 3000 --hunt $O/final_h.pt > $O/final_probe.log 2>&1
echo "=== RESULTS ==="
for f in $O/final_bench.json $O/final_bench_lstm.json $O/final_cap.json \
         $O/final_cap_lstm.json $O/final_cap_deep.json; do
  echo "--- $(basename $f)"; cat $f; echo
done
grep -h "\"probe\"" $O/final_probe.log | tail -1
