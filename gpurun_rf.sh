#!/bin/bash
cd /root/repo
python -m pytest tests/test_gpu_kernels.py -q -m gpu -x 2>&1 | tail -1
timeout 600 python tools/bench_rf.py --trees 300 2>/dev/null | tail -1
timeout 600 python tools/bench_rf.py --trees 300 2>/dev/null | tail -1
python bench.py --steps 50 --warmup 8 2>/dev/null | python -c "import json,sys; d=json.load(sys.stdin); print('11M', round(d['value'],1))"
python bench.py --rows 1375000 --steps 60 --warmup 10 2>/dev/null | python -c "import json,sys; d=json.load(sys.stdin); print('1.375M', round(d['value'],1))"
