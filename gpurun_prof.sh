#!/bin/bash
cd /tmp && export TMPDIR=/tmp
cd /root/repo
mkdir -p gpurun_out/prof_r02b
rocprofv3 --kernel-trace --stats -d gpurun_out/prof_r02b -o full -- \
  python bench.py --steps 30 --warmup 5 > gpurun_out/prof_r02b/full.log 2>&1
tail -2 gpurun_out/prof_r02b/full.log | head -1
rocprofv3 --kernel-trace --stats -d gpurun_out/prof_r02b -o small -- \
  python bench.py --rows 1375000 --steps 60 --warmup 10 > gpurun_out/prof_r02b/small.log 2>&1
grep -o '"value": [0-9.]*' gpurun_out/prof_r02b/small.log | head -1
ls gpurun_out/prof_r02b/
