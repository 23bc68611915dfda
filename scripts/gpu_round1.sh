#!/bin/bash
# Round-1 GPU evidence pass: tests, bench, rocprof. Run via gpurun from repo root.
set -x
mkdir -p gpurun_out
R=$PWD

echo "== gpu tests ==" > gpurun_out/summary.txt
timeout 300 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest_gpu exit=$?" >> gpurun_out/summary.txt
tail -5 gpurun_out/pytest_gpu.log >> gpurun_out/summary.txt

echo "== bench repo config (1 GPU) ==" >> gpurun_out/summary.txt
timeout 300 python bench.py --config repo --steps 50 --warmup 10 > gpurun_out/bench_repo.json 2> gpurun_out/bench_repo.err
echo "bench_repo exit=$?" >> gpurun_out/summary.txt
cat gpurun_out/bench_repo.json >> gpurun_out/summary.txt

echo "== bench predict config ==" >> gpurun_out/summary.txt
timeout 300 python bench.py --config predict --steps 200 --warmup 20 > gpurun_out/bench_predict.json 2> gpurun_out/bench_predict.err
echo "bench_predict exit=$?" >> gpurun_out/summary.txt
cat gpurun_out/bench_predict.json >> gpurun_out/summary.txt

echo "== bench stress config ==" >> gpurun_out/summary.txt
timeout 420 python bench.py --config stress --steps 10 --warmup 3 > gpurun_out/bench_stress.json 2> gpurun_out/bench_stress.err
echo "bench_stress exit=$?" >> gpurun_out/summary.txt
cat gpurun_out/bench_stress.json >> gpurun_out/summary.txt

echo "== rocprof kernel stats (repo config) ==" >> gpurun_out/summary.txt
cd /tmp && export TMPDIR=/tmp
timeout 420 rocprofv3 --kernel-trace --stats -d $R/gpurun_out/prof -o repo_train -- python $R/bench.py --config repo --steps 20 --warmup 5 > $R/gpurun_out/rocprof.log 2>&1
echo "rocprof exit=$?" >> $R/gpurun_out/summary.txt
find $R/gpurun_out/prof -name '*stats*' >> $R/gpurun_out/summary.txt
cat $R/gpurun_out/summary.txt
