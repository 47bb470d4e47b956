set -x
python - <<'PY'
import sys, torch, time; sys.path.insert(0, '.')
from lstm_tensorspark_amd.config import get_preset
from lstm_tensorspark_amd.trainer import Trainer
for dt in ("bf16", "fp8"):
    cfg = get_preset("lm-50k"); cfg.train.log_interval = 0; cfg.train.ckpt_dir = ""
    cfg.train.dtype = dt
    t = Trainer(cfg, device="cuda:0")
    b = t.data.batch(0)
    for _ in range(2): t.train_step(b)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(4): t.train_step(b)
    torch.cuda.synchronize()
    ms = (time.perf_counter()-t0)/4*1e3
    print(f"lm-50k {dt}: {ms:.1f} ms/step ({cfg.train.batch_size/ms*1000:.0f} seq/s)")
    del t; torch.cuda.empty_cache()
PY
timeout 300 python tools/decode_bench.py --B 256 --steps 64 2>&1 | tail -1
timeout 300 python train.py --preset lm-50k --set train.batch_size=128 \
  --set train.steps=60 --set train.log_interval=20 --set data.task=lm_assoc \
  --set train.lr=0.0003 --set train.dtype=fp8 2>&1 | grep '"step"'
