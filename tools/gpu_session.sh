set -x
mkdir -p gpurun_out
timeout 500 python train.py --preset lm-50k --set train.batch_size=128 \
  --set train.steps=400 --set train.log_interval=80 --set data.task=lm_assoc \
  --set train.lr=0.0003 2>&1 | grep '"step"' | tee gpurun_out/curve_lm_long.jsonl
timeout 300 python train.py --preset mi355x-1gpu --set model.hidden_size=16384 \
  --set model.input_size=16384 --set train.batch_size=512 --set train.seq_len=64 \
  --set train.steps=4 --set train.log_interval=2 2>&1 | grep '"step"\|Error' | tail -2
rocm-smi --showmeminfo vram 2>/dev/null | grep Used
