set -x
python - <<'PY'
import __graft_entry__
__graft_entry__.smoke()
PY
python -m pytest tests -m gpu -q 2>&1 | tail -1
timeout 300 python tools/kernel_bench.py --B 4096 --H 1024 --T 16 --iters 20 2>&1 | grep -E "best|seq_fwd p|seq_bwd|layer" | head -5
timeout 300 python bench.py --steps 10 --warmup 2 2>&1 | tail -1
