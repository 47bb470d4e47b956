set -x
for b in 256 512 1024; do
  timeout 400 python bench.py --gpus 1 --steps 6 --warmup 2 --preset bidir-4l --batch $b 2>&1 | tail -1
done
for b in 128 256 512; do
  timeout 400 python bench.py --gpus 1 --steps 6 --warmup 2 --preset lm-50k --batch $b 2>&1 | tail -1
done
