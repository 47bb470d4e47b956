set -x
mkdir -p gpurun_out
timeout 600 python train.py --preset lm-50k --set train.batch_size=128 \
  --set train.steps=100 --set train.log_interval=10 --set data.task=lm_assoc \
  --set train.lr=0.0003 2>&1 | grep '"step"' | tee gpurun_out/curve_lm.jsonl
