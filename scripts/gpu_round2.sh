#!/usr/bin/env bash
# Round-2 verification recipe (run on an MI355X box from the repo root).
# Reproduces every measurement cited in profiles/r02_*.md.
set -x

# 1) full GPU suite (82 tests) + build smoke
python -m pytest tests -m gpu -q
python -c "from __graft_entry__ import smoke; smoke(); print('SMOKE OK')"

# 2) the three benchmark configs (BASELINE.json)
python bench.py --config repo    --steps 40  --warmup 10   # ~1.05M seq/s
python bench.py --config stress  --steps 6   --warmup 2    # ~7.2k seq/s
python bench.py --config predict --steps 300 --warmup 30   # p50 ~0.21 ms

# 3) kernel stats + PMC evidence (counters in their OWN run; never combine
#    --pmc with trace domains)
cd /tmp && export TMPDIR=/tmp && cd - > /dev/null
rocprofv3 --kernel-trace --stats -d /tmp/prof -o r2 -- \
    python bench.py --config repo --steps 10 --warmup 5
rocprofv3 --pmc SQ_BUSY_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY \
    SQ_WAIT_INST_LDS SQ_LDS_BANK_CONFLICT SQ_ACTIVE_INST_VALU \
    SQ_ACTIVE_INST_LDS SQ_INSTS_VALU_MFMA_BF16 \
    --kernel-include-regex "gru_" -d /tmp/pmc -o r2p -- \
    python bench.py --config repo --steps 5 --warmup 3
python scripts/pmc_extract.py "/tmp/pmc/**/*_results.db" pmc_summary.txt

# 4) kernel microbenchmarks (per-kernel CUDA-event timing)
python scripts/kernel_micro.py --iters 50                  # fwd/bwd v3
python scripts/kernel_micro.py --what outer --B 8192 --iters 30

# 5) multi-epoch quality + CPU/GPU training equivalence
python -m fmda_amd.train --epochs 25 --hidden 32 --layers 1 --rows 3980 \
    --window 30 --batch 16 --device cuda --log-file quality.jsonl
python -m pytest tests/test_gpu_train_equivalence.py -q

# 6) host sanitizer run over the native checkpoint serializer (CPU-only;
#    also runs in the build container)
python scripts/asan_checkpoint.py
