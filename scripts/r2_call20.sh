#!/bin/bash
# Subsystem isolation matrix: which async overlap produces the bad grads?
# Metric: grad_skip_count over 4000 full-speed steps (no per-step sync).
set -x
mkdir -p gpurun_out
O=$PWD/gpurun_out
run() {
  name=$1; shift
  timeout 300 env "$@" python scripts/nan_probe.py --steps 4000 \
      --hunt $O/r20_$name.pt > $O/r20_$name.log 2>&1
  echo "=== $name rc=$?"
  grep -h '"probe"' $O/r20_$name.log | tail -1 | python3 -c "
import json,sys
d=json.loads(sys.stdin.read())
f=d['final']
print(' skips=', f.get('grad_skip_count'), ' first_bad=', d.get('first_nonfinite_step'), ' fps=', round(f.get('fps',0)))"
}
run base DUMMY=1
run nocopyovl IMPALA_NO_COPY_OVERLAP=1
run nopub IMPALA_NO_ASYNC_PUBLISH=1
run noboth IMPALA_NO_COPY_OVERLAP=1 IMPALA_NO_ASYNC_PUBLISH=1
