#!/bin/bash
cd /root/repo
run() {
  local tag="$1" rows="$2"; shift 2
  env "$@" python bench.py --rows $rows --steps 50 --warmup 8 2>/dev/null \
    | python -c "import json,sys; d=json.load(sys.stdin); print('$tag', round(d['value'],1),'trees/s', round(d['ms_per_step']*1000,1),'us')"
}
run 11M_i16dense 11000000 YDFA_I16_DENSE=1
run 1.375M_i16dense 1375000 YDFA_I16_DENSE=1
timeout 420 python tools/bench_rf.py --trees 150 2>/dev/null | tail -1
