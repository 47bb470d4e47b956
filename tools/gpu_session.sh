set -x
mkdir -p gpurun_out
# convergence evidence: loss curves on real shapes
timeout 500 python train.py --preset mi355x-1gpu --set train.batch_size=4096 \
  --set train.steps=60 --set train.log_interval=10 --set data.task=copy \
  --set model.input_size=1024 2>&1 | grep -E '"step"' | tee gpurun_out/curve_copy.jsonl
timeout 500 python train.py --preset lm-50k --set train.batch_size=128 \
  --set train.steps=30 --set train.log_interval=5 2>&1 | grep -E '"step"' | tee gpurun_out/curve_lm.jsonl
