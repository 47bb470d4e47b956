set -x
mkdir -p gpurun_out
R=$GRAFT_REPO_ROOT
timeout 420 python tools/kernel_bench.py --B 4096 --H 1024 --T 32 --iters 50 2>&1 | tee gpurun_out/kbench_bm128.log
cd /tmp && export TMPDIR=/tmp
timeout 420 rocprofv3 --kernel-trace --stats -d $R/gpurun_out/prof2 -- python $R/bench.py --steps 10 --warmup 3 > $R/gpurun_out/bench_prof2.log 2>&1
tail -2 $R/gpurun_out/bench_prof2.log
