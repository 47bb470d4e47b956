set -x
python -m pytest tests/test_gpu_kernels.py -m gpu -q 2>&1 | tail -1
for b in 4096 6144 8192; do
  timeout 300 python bench.py --gpus 1 --steps 20 --warmup 4 --batch $b 2>&1 | tail -1
done
