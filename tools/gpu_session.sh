set -x
timeout 420 python tools/kernel_bench.py --B 4096 --H 1024 --T 32 --iters 50 2>&1 | grep -E "variant=(2|32|33|40|41)|best|RAW|seq_fwd p"
for v in 32 96 160 33; do timeout 200 python tools/pmc_rec_fwd.py --variant $v --raw --iters 50 2>&1 | tail -1; done
timeout 300 python bench.py --gpus 1 --steps 30 --warmup 5 2>&1 | tail -1
