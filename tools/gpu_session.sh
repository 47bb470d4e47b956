set -x
python -m pytest tests/test_gpu_kernels.py -m gpu -q 2>&1 | tail -1
timeout 400 python bench.py --gpus 1 --steps 12 --warmup 3 2>&1 | tail -1
