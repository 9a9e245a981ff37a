#!/bin/bash
cd /root/repo
python -m pytest tests/test_gpu_kernels.py -q -m gpu 2>&1 | tail -1
echo "=== RF bench ==="
timeout 420 python tools/bench_rf.py --trees 150 2>&1 | tail -2
run() {
  local tag="$1" rows="$2"
  python bench.py --rows $rows --steps 50 --warmup 8 2>/dev/null \
    | python -c "import json,sys; d=json.load(sys.stdin); print('$tag', round(d['value'],1),'trees/s')"
}
run 11M 11000000
run 1.375M 1375000
