set -x
timeout 400 python bench.py --gpus 1 --steps 10 --warmup 3 2>&1 | tail -1
timeout 400 python bench.py --gpus 1 --steps 6 --warmup 2 --preset bidir-4l 2>&1 | tail -1
