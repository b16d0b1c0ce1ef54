#!/bin/bash
# Round-3 opening GPU ladder (run first):
#   /usr/local/graft/bin/gpurun --timeout 1500 -- 'bash scripts/round3_opening.sh'
set -x
mkdir -p gpurun_out
R=$PWD
O=$PWD/gpurun_out

# 0. probes + suite: confirm the round-2 frozen state on this round's image
./torched_impala_amd/csrc/deep_probe 2>&1 | tail -3 | tee $O/r3o_probe.log
timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -3 | tee $O/r3o_tests.log

# 1. baselines
timeout 600 python bench.py --steps 200 --warmup 30 > $O/r3o_bench.json 2>/dev/null
timeout 600 python bench.py --steps 200 --warmup 30 --use-lstm > $O/r3o_bench_lstm.json 2>/dev/null
timeout 300 python scripts/learner_capacity.py --graphs --steps 150 --warmup 30 > $O/r3o_cap.json 2>/dev/null
timeout 300 python scripts/learner_capacity.py --graphs --use-lstm --steps 150 --warmup 30 > $O/r3o_cap_lstm.json 2>/dev/null

# 2. LSTM PMC (the round-3 headline lever: lstm_seq_fwd/bwd ~390 us/step)
cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY \
    SQ_ACTIVE_INST_ANY --output-format csv -d /tmp/r3pmc -o pmc -- \
    python $R/scripts/learner_capacity.py --graphs --use-lstm --steps 30 \
    --warmup 10 > $O/r3o_pmc.log 2>&1
find /tmp/r3pmc -name "*.csv" -exec cp {} $O/ \;
cd $R

# 3. hipBLASLt bf16-under-graph standalone repro attempt
timeout 400 python scripts/blaslt_bf16_graph_repro.py --iters 20000 \
    > $O/r3o_blaslt.log 2>&1
echo "BLASLT_RC=$? (0 = minimal form does not reproduce)"
tail -2 $O/r3o_blaslt.log

# 4. stability spot-check
timeout 260 python scripts/nan_probe.py --steps 3000 --hunt $O/r3o_h.pt > $O/r3o_probe_nan.log 2>&1
grep -h "\"probe\"" $O/r3o_probe_nan.log | tail -1

echo "=== JSON ==="
for f in $O/r3o_bench.json $O/r3o_bench_lstm.json $O/r3o_cap.json $O/r3o_cap_lstm.json; do
  echo "--- $f"; cat $f 2>/dev/null
done
