#!/bin/bash
cd /root/repo
echo "=== full GPU pytest ==="
python -m pytest tests/ -q -m gpu 2>&1 | tail -1
echo "=== smoke ==="
python -c "import __graft_entry__ as g; g.smoke(); print('smoke OK')" 2>&1 | tail -1
echo "=== bench driver-style 300 steps ==="
python bench.py 2>/dev/null
echo "=== strong-scaling ladder (1 GPU) ==="
for r in 5500000 2750000 1375000; do
  python bench.py --rows $r --steps 60 --warmup 10 2>/dev/null | python -c "import json,sys; d=json.load(sys.stdin); print(d['config']['rows'], round(d['value'],1), 'trees/s')"
done
