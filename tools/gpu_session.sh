set -x
python -m pytest tests/test_gpu_kernels.py tests/test_gpu_train.py -m gpu -q 2>&1 | tail -1
for b in 1 4 16; do
  timeout 200 python tools/decode_bench.py --B $b --steps 256 2>&1 | tail -1
done
