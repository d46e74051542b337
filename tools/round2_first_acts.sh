#!/bin/bash
# Round-2 opening GPU validations (DESIGN.md §7 "Round-2 first acts").
# Run each block through gpurun; budget ~10 GPU-minutes total.

# 1. Which scan did the driver's BENCH_r01 use, and what does refine give?
#    (the adaptive default falls back if recall < 0.95 — confirm the branch
#    and sweep R for the recall/QPS knee)
for R in 64 128 256 512; do
  timeout 300 python bench.py --steps 10 --warmup 3 --refine $R \
    --skip-cpu-baseline 2>&1 | grep -E "recall|perf:|\"value\""
done
# exact path reference on the same box:
timeout 300 python bench.py --steps 10 --warmup 3 --refine 0 \
  --skip-cpu-baseline 2>&1 | grep -E "recall|perf:"

# 2. int8 storage true recall at 10M (post grid-wrap fix)
timeout 300 python bench.py --qtype int8 --steps 6 --warmup 2 \
  --skip-cpu-baseline 2>&1 | grep -E "recall|perf:"

# 3. f16 line steady-state (its scan is the next asm-scheduling target)
timeout 300 python bench.py --qtype f16 --steps 10 --warmup 3 \
  --skip-cpu-baseline 2>&1 | grep -E "recall|perf:"

# 4. full suites at HEAD
python -m pytest tests/ -q -m gpu | tail -2
python -c "import __graft_entry__ as g; g.smoke()"
