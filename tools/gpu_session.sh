set -x
timeout 500 python -m torch.distributed.run --nnodes=1 --nproc-per-node 1 \
  --master-addr 127.0.0.1 --master-port 29601 bench.py --gpus 1 --steps 8 --warmup 2 2>&1 | tail -1
timeout 400 python bench.py --gpus 1 --steps 8 --warmup 2 --dtype fp8 2>&1 | tail -1
