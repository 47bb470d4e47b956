set -x
mkdir -p gpurun_out
R=$GRAFT_REPO_ROOT
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats -d $R/gpurun_out/prof_lm -- python $R/bench.py --steps 3 --warmup 1 --preset lm-50k > $R/gpurun_out/prof_lm.log 2>&1
timeout 300 rocprofv3 --kernel-trace --stats -d $R/gpurun_out/prof_bd -- python $R/bench.py --steps 3 --warmup 1 --preset bidir-4l > $R/gpurun_out/prof_bd.log 2>&1
echo done
