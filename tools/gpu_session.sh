set -x
mkdir -p gpurun_out
R=$GRAFT_REPO_ROOT
python -m pytest tests -m gpu -q 2>&1 | tail -30 | tee gpurun_out/pytest_gpu.log
timeout 300 python bench.py --gpus 1 --steps 30 --warmup 5 2>&1 | tee gpurun_out/bench_1gpu.json
cd /tmp && export TMPDIR=/tmp
timeout 420 rocprofv3 --kernel-trace --stats -d $R/gpurun_out/prof -- python $R/bench.py --steps 10 --warmup 3 > $R/gpurun_out/bench_prof.log 2>&1
tail -5 $R/gpurun_out/bench_prof.log
find $R/gpurun_out/prof -type f | head
