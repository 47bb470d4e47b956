set -x
mkdir -p gpurun_out
R=$GRAFT_REPO_ROOT
python - <<'PY'
import __graft_entry__
__graft_entry__.smoke()
PY
python -m pytest tests -m gpu -q 2>&1 | tail -1 | tee gpurun_out/final_pytest.log
timeout 400 python bench.py --gpus 1 --steps 15 --warmup 3 2>&1 | tail -1 | tee gpurun_out/final_bench.json
timeout 400 python bench.py --gpus 1 --steps 5 --warmup 2 --preset bidir-4l 2>&1 | tail -1 | tee -a gpurun_out/final_bench.json
timeout 400 python bench.py --gpus 1 --steps 5 --warmup 2 --preset lm-50k 2>&1 | tail -1 | tee -a gpurun_out/final_bench.json
for b in 1 4 256 512; do timeout 200 python tools/decode_bench.py --B $b --steps 192 2>&1 | tail -1; done | tee gpurun_out/decode_sweep.log
cd /tmp && export TMPDIR=/tmp
timeout 420 rocprofv3 --kernel-trace --stats -d $R/gpurun_out/proff2 -- python $R/bench.py --steps 5 --warmup 2 > $R/gpurun_out/final_prof.log 2>&1
grep -o '"value": [0-9.]*' $R/gpurun_out/final_prof.log | head -1
