#!/bin/bash
# Round-2 GPU call 4: split-K linear + direct-slot-write actors + ramp
# barrier validation; rocprof CSV attribution of fp32-vs-MFMA capacity;
# actor supply curve.
set -x
mkdir -p gpurun_out
R=$PWD
O=$PWD/gpurun_out

timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -6 \
    | tee $O/r4_gpu_tests.log

timeout 600 python bench.py --steps 40 --warmup 12 \
    > $O/r4_bench.json 2> $O/r4_bench.err
echo "BENCH_RC=$?"
timeout 600 python bench.py --steps 40 --warmup 12 --use-lstm \
    > $O/r4_bench_lstm.json 2> $O/r4_bench_lstm.err
echo "BENCH_LSTM_RC=$?"

for A in 16 64; do
  timeout 300 python scripts/actor_probe.py --jit native-bf16 --actors $A \
      --seconds 8 > $O/r4_actor$A.json 2> $O/r4_actor$A.err
done

cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats --output-format csv \
    -d /tmp/p_fp32 -o fp32 -- \
    python $R/scripts/learner_capacity.py --steps 60 --warmup 15 \
    > $O/r4_cap_fp32_prof.log 2>&1
IMPALA_SHALLOW_MFMA=1 timeout 600 rocprofv3 --kernel-trace --stats \
    --output-format csv -d /tmp/p_mfma -o mfma -- \
    python $R/scripts/learner_capacity.py --steps 60 --warmup 15 \
    > $O/r4_cap_mfma_prof.log 2>&1
find /tmp/p_fp32 /tmp/p_mfma -name "*kernel_stats*" | head
find /tmp/p_fp32 -name "*kernel_stats*" -exec cp {} $O/r4_fp32_kernel_stats.csv \;
find /tmp/p_mfma -name "*kernel_stats*" -exec cp {} $O/r4_mfma_kernel_stats.csv \;
cd $R

du -sh $O
echo "=== JSON dump ==="
for f in $O/r4_bench.json $O/r4_bench_lstm.json $O/r4_actor16.json \
         $O/r4_actor64.json; do
  echo "--- $f"; cat "$f" 2>/dev/null
done
tail -n 3 $O/r4_bench.err
grep -h "ms_per_step" $O/r4_cap_fp32_prof.log $O/r4_cap_mfma_prof.log
