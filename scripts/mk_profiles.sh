#!/bin/bash
# Regenerates profiles/*_final.md: rocprofv3 kernel traces for all four
# model configs, parsed from the results.db into per-kernel tables.
# Run on a GPU box (gpurun -- bash scripts/mk_profiles.sh).
export TMPDIR=/tmp
cd /root/repo
rm -rf gpurun_out/*
for m in alexnet cifar10_quick googlenet lrcn; do
  rocprofv3 --kernel-trace --stats -d gpurun_out/prof_$m -o $m -- \
    python bench.py --gpus 1 --steps 20 --warmup 6 --model $m \
    > gpurun_out/prof_$m.log 2>&1
  grep -o '"ms_per_step": [0-9.]*' gpurun_out/prof_$m.log | tail -1
done
python scripts/stats2md.py
rm -rf gpurun_out/prof_*/   # keep only the md summaries + logs (64MiB cap)
