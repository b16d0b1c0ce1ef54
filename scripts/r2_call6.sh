#!/bin/bash
# Round-2 GPU call 6: fixed suite; actor pinning study; PMC + kernel
# stats of the graph-mode MFMA step; deep graph-mode attempt.
set -x
mkdir -p gpurun_out
R=$PWD
O=$PWD/gpurun_out

timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -5 \
    | tee $O/r6_gpu_tests.log

for CFG in "32 --pin" "64 --pin" "128 --pin" "64"; do
  set -- $CFG
  A=$1; PIN=${2:-}
  timeout 300 python scripts/actor_probe.py --jit native-bf16 \
      --actors $A $PIN --seconds 8 \
      > $O/r6_actor${A}${PIN:+_pin}.json 2> /dev/null
done

# deep big config through the graph path (compat previously unproven)
timeout 600 python scripts/learner_capacity.py --model deep --use-lstm \
    --batch-size 512 --unroll-length 100 --steps 8 --warmup 3 --graphs \
    > $O/r6_deep_graphs.json 2> $O/r6_deep_graphs.err
echo "DEEP_GRAPHS_RC=$?"

cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats --output-format csv \
    -d /tmp/p_g -o g -- \
    python $R/scripts/learner_capacity.py --graphs --steps 100 --warmup 20 \
    > $O/r6_capg_prof.log 2>&1
find /tmp/p_g -name "*kernel_stats*" -exec cp {} $O/r6_graph_kernel_stats.csv \;
timeout 600 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY \
    SQ_ACTIVE_INST_ANY --output-format csv -d /tmp/p_pmc -o pmc -- \
    python $R/scripts/learner_capacity.py --graphs --steps 30 --warmup 10 \
    > $O/r6_pmc.log 2>&1
echo "PMC_RC=$?"
ls /tmp/p_pmc
find /tmp/p_pmc -name "*counter*" -o -name "*pmc*" | head -5
for f in $(find /tmp/p_pmc -name "*.csv" | head -3); do
  cp "$f" $O/r6_pmc_$(basename $f)
done
cd $R
du -sh $O
echo "=== JSON dump ==="
for f in $O/r6_actor*.json $O/r6_deep_graphs.json; do
  echo "--- $f"; cat "$f" 2>/dev/null
done
tail -n 5 $O/r6_deep_graphs.err 2>/dev/null
