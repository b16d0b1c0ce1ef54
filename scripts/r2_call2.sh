#!/bin/bash
# Round-2 GPU call 2: full suite + headline benches + MFMA-capacity
# rocprof attribution + flake repro.
set -x
mkdir -p gpurun_out
R=$PWD
O=$PWD/gpurun_out

timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -8 \
    | tee $O/r2c2_gpu_tests.log

timeout 600 python bench.py --steps 40 --warmup 12 \
    > $O/r2c2_bench.json 2> $O/r2c2_bench.err
timeout 600 python bench.py --steps 40 --warmup 12 --use-lstm \
    > $O/r2c2_bench_lstm.json 2> $O/r2c2_bench_lstm.err

timeout 300 python scripts/actor_probe.py --jit native-bf16 --actors 32 \
    --seconds 8 > $O/r2c2_actor32.json 2> $O/r2c2_actor32.err

cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats -d $O/prof_cap_fp32 -- \
    python $R/scripts/learner_capacity.py --steps 60 --warmup 15 \
    > $O/r2c2_cap_fp32_prof.log 2>&1
IMPALA_SHALLOW_MFMA=1 timeout 600 rocprofv3 --kernel-trace --stats \
    -d $O/prof_cap_mfma -- \
    python $R/scripts/learner_capacity.py --steps 60 --warmup 15 \
    > $O/r2c2_cap_mfma_prof.log 2>&1
cd $R

timeout 300 python scripts/graph_flake_repro.py --iters 12 \
    > $O/r2c2_flake_nodestroy.log 2>&1 \
    || echo "FLAKE_NODESTROY rc=$?" | tee -a $O/r2c2_flake_nodestroy.log
timeout 300 python scripts/graph_flake_repro.py --iters 12 --teardown \
    > $O/r2c2_flake_destroy.log 2>&1 \
    || echo "FLAKE_DESTROY rc=$?" | tee -a $O/r2c2_flake_destroy.log

echo "=== summary ==="
tail -2 $O/r2c2_gpu_tests.log
tail -1 $O/r2c2_bench.json $O/r2c2_bench_lstm.json $O/r2c2_actor32.json
grep -h "ms_per_step\|frames_per_sec" $O/r2c2_cap_fp32_prof.log \
    $O/r2c2_cap_mfma_prof.log | tail -4
tail -1 $O/r2c2_flake_nodestroy.log $O/r2c2_flake_destroy.log
