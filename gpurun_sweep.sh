#!/bin/bash
cd /root/repo
python -m pytest tests/test_gpu_kernels.py -q -m gpu -x 2>&1 | tail -1
run() {
  local tag="$1" rows="$2"
  python bench.py --rows $rows --steps 60 --warmup 10 2>/dev/null \
    | python -c "import json,sys; d=json.load(sys.stdin); print('$tag', round(d['value'],1),'trees/s', round(d['ms_per_step']*1000,1),'us')"
}
run 11M_psum 11000000
run 1.375M_psum 1375000
run 1.375M_psum_b 1375000
