set -x
mkdir -p gpurun_out
R=$GRAFT_REPO_ROOT
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --pmc MfmaUtil VALUBusy OccupancyPercent -d $R/gpurun_out/pmcu1 --output-format csv -- python $R/tools/pmc_rec_fwd.py --variant 2 --iters 30 > $R/gpurun_out/pmcu1.log 2>&1 || tail -3 $R/gpurun_out/pmcu1.log
timeout 300 rocprofv3 --pmc MfmaUtil VALUBusy OccupancyPercent -d $R/gpurun_out/pmcu2 --output-format csv -- python $R/tools/pmc_rec_fwd.py --variant 32 --iters 30 > $R/gpurun_out/pmcu2.log 2>&1 || tail -3 $R/gpurun_out/pmcu2.log
timeout 400 rocprofv3 --pmc MfmaUtil VALUBusy OccupancyPercent -d $R/gpurun_out/pmcu3 --output-format csv -- python $R/bench.py --steps 3 --warmup 1 --batch 4096 > $R/gpurun_out/pmcu3.log 2>&1 || tail -3 $R/gpurun_out/pmcu3.log
find $R/gpurun_out/pmcu* -name "*counter*" | head
