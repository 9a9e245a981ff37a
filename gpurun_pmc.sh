#!/bin/bash
cd /tmp && export TMPDIR=/tmp
cd /root/repo
mkdir -p gpurun_out/pmc_chunks
rocprofv3 --pmc TCC_HIT_sum TCC_MISS_sum -d gpurun_out/pmc_chunks -o new -- \
  python bench.py --rows 1375000 --steps 20 --warmup 4 > gpurun_out/pmc_chunks/new.log 2>&1
echo "new rc=$?"
YDFA_HIST_MAX_BLOCKS=2048 rocprofv3 --pmc TCC_HIT_sum TCC_MISS_sum -d gpurun_out/pmc_chunks -o old -- \
  python bench.py --rows 1375000 --steps 20 --warmup 4 > gpurun_out/pmc_chunks/old.log 2>&1
echo "old rc=$?"
ls gpurun_out/pmc_chunks/
