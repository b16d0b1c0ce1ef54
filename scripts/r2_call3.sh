#!/bin/bash
# Round-2 GPU call 3: new grid-stride dW validation, headline benches
# (echoed to stdout — call 2's artifacts were lost to the 64 MiB cap),
# rocprof capacity A/B with traces kept in /tmp, deep full-step, flake
# repro.
set -x
mkdir -p gpurun_out
R=$PWD
O=$PWD/gpurun_out

./torched_impala_amd/csrc/deep_probe 2>&1 | tee $O/r3_probe.log | tail -30
PROBE_RC=${PIPESTATUS[0]}
echo "PROBE_RC=$PROBE_RC"
./torched_impala_amd/csrc/deep_probe bench 2>&1 | tee $O/r3_probe_bench.log

timeout 600 python bench.py --steps 40 --warmup 12 \
    > $O/r3_bench.json 2> $O/r3_bench.err
echo "BENCH_RC=$?"
timeout 600 python bench.py --steps 40 --warmup 12 --use-lstm \
    > $O/r3_bench_lstm.json 2> $O/r3_bench_lstm.err
echo "BENCH_LSTM_RC=$?"

timeout 300 python scripts/actor_probe.py --jit native-bf16 --actors 32 \
    --seconds 8 > $O/r3_actor32.json 2> $O/r3_actor32.err

# deep full-step capacity on the custom kernels,
# headline-size big config, fp32 and bf16 autocast
timeout 600 python scripts/learner_capacity.py --model deep --use-lstm \
    --batch-size 512 --unroll-length 100 --steps 8 --warmup 3 \
    > $O/r3_deep_cap_fp32.json 2> $O/r3_deep_cap_fp32.err
echo "DEEP_FP32_RC=$?"
timeout 600 python scripts/learner_capacity.py --model deep --use-lstm \
    --batch-size 512 --unroll-length 100 --steps 8 --warmup 3 \
    --compute-dtype bf16 \
    > $O/r3_deep_cap_bf16.json 2> $O/r3_deep_cap_bf16.err
echo "DEEP_BF16_RC=$?"

# rocprof capacity A/B — traces stay in /tmp; only kernel_stats come back
cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats -d /tmp/prof_fp32 -- \
    python $R/scripts/learner_capacity.py --steps 60 --warmup 15 \
    > $O/r3_cap_fp32_prof.log 2>&1
IMPALA_SHALLOW_MFMA=1 timeout 600 rocprofv3 --kernel-trace --stats \
    -d /tmp/prof_mfma -- \
    python $R/scripts/learner_capacity.py --steps 60 --warmup 15 \
    > $O/r3_cap_mfma_prof.log 2>&1
find /tmp/prof_fp32 /tmp/prof_mfma -name "*kernel_stats*" \
    -exec cp {} $O/ \; 2>/dev/null
for f in /tmp/prof_fp32/*/*kernel_stats.csv; do cp "$f" $O/r3_fp32_kernel_stats.csv; done
for f in /tmp/prof_mfma/*/*kernel_stats.csv; do cp "$f" $O/r3_mfma_kernel_stats.csv; done
cd $R

timeout 300 python scripts/graph_flake_repro.py --iters 12 \
    > $O/r3_flake_nodestroy.log 2>&1
echo "FLAKE_NODESTROY_RC=$?"
timeout 300 python scripts/graph_flake_repro.py --iters 12 --teardown \
    > $O/r3_flake_destroy.log 2>&1
echo "FLAKE_DESTROY_RC=$?"

du -sh $O
echo "=== JSON dump ==="
for f in $O/r3_bench.json $O/r3_bench_lstm.json $O/r3_actor32.json \
         $O/r3_deep_cap_fp32.json $O/r3_deep_cap_bf16.json; do
  echo "--- $f"; cat "$f" 2>/dev/null
done
echo "=== bench stderr tails ==="
tail -n 3 $O/r3_bench.err $O/r3_bench_lstm.err 2>/dev/null
grep -h "ms_per_step" $O/r3_cap_fp32_prof.log $O/r3_cap_mfma_prof.log
tail -n 1 $O/r3_flake_nodestroy.log $O/r3_flake_destroy.log
