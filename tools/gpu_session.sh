set -x
mkdir -p gpurun_out
for b in 1 16 64 256 512; do
  timeout 200 python tools/decode_bench.py --B $b --steps 64 2>&1 | tail -1
done | tee gpurun_out/decode_sweep.log
timeout 400 python bench.py --gpus 1 --steps 6 --warmup 2 --preset bidir-4l --dtype fp8 2>&1 | tail -1
