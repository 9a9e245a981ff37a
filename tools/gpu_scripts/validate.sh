#!/bin/bash
cd /root/repo
echo "=== full GPU pytest ==="
python -m pytest tests/ -q -m gpu 2>&1 | tail -1
echo "=== smoke ==="
python -c "import __graft_entry__ as g; g.smoke(); print('smoke OK')" 2>&1 | tail -1
echo "=== bench driver-style 300 steps ==="
python bench.py 2>/dev/null
echo "=== RF 300 trees ==="
timeout 600 python tools/bench_rf.py --trees 300 2>/dev/null | tail -1 | python -c "import json,sys; d=json.load(sys.stdin); print('RF', round(d['value'],1),'trees/s')"
