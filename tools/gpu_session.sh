set -x
python -m pytest tests/test_gpu_kernels.py tests/test_gpu_train.py -m gpu -q 2>&1 | tail -2
timeout 300 python bench.py --gpus 1 --steps 30 --warmup 5 2>&1 | tail -1
