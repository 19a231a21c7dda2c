#!/usr/bin/env bash
# Round-end GPU validation bundle (run on the box via gpurun):
#   pytest -m gpu, kernel microbench, stem microbench, e2e bench, rocprof digest.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp
timeout 300 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest_rc=$?" >> gpurun_out/pytest_gpu.log
tail -3 gpurun_out/pytest_gpu.log
timeout 200 python benchmarks/kernel_bench.py > gpurun_out/kernel_bench.log 2>&1
timeout 200 python benchmarks/stem_bench.py 2048 > gpurun_out/stem_bench_2048.log 2>&1
timeout 420 python bench.py --steps 10 --warmup 3 > gpurun_out/bench_b2048.json 2>&1
tail -1 gpurun_out/bench_b2048.json
cd /tmp
timeout 420 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof -o r02f -- python /root/repo/bench.py --steps 4 --warmup 2 > /root/repo/gpurun_out/bench_prof.log 2>&1
cd /root/repo
python scripts/prof_summary.py gpurun_out/prof/*r02f*.db 45 > gpurun_out/r02_final_digest.md 2>&1 || ls gpurun_out/prof
rm -rf gpurun_out/prof
echo DONE
