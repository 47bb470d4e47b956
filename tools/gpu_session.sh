set -x
mkdir -p gpurun_out
for dt in bf16 fp8; do
  timeout 400 python train.py --preset lm-50k --set train.batch_size=128 \
    --set train.steps=150 --set train.log_interval=30 --set data.task=lm_assoc \
    --set train.lr=0.0003 --set train.dtype=$dt 2>&1 | grep '"step"' | sed "s/^/[$dt] /"
done | tee gpurun_out/curve_lm_fp8_ab.log
