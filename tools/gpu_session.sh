set -x
python -m pytest tests/test_gpu_kernels.py -m gpu -q 2>&1 | tail -1
python - <<'PY'
import sys, torch, time; sys.path.insert(0, '.')
from lstm_tensorspark_amd.config import get_preset
from lstm_tensorspark_amd.trainer import Trainer
for dt in ("bf16", "fp8"):
    cfg = get_preset("mi355x-1gpu"); cfg.train.log_interval = 0; cfg.train.ckpt_dir = ""
    cfg.train.dtype = dt
    t = Trainer(cfg, device="cuda:0")
    b = t.data.batch(0)
    for _ in range(3): t.train_step(b)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(6): t.train_step(b)
    torch.cuda.synchronize()
    ms = (time.perf_counter()-t0)/6*1e3
    print(f"{dt}: {ms:.1f} ms/step ({cfg.train.batch_size/ms*1000:.0f} seq/s)")
    del t; torch.cuda.empty_cache()
PY
timeout 300 python train.py --preset mi355x-1gpu --set train.batch_size=4096 \
  --set train.steps=100 --set train.log_interval=50 --set data.task=copy \
  --set model.input_size=1024 --set train.lr=0.4 --set train.dtype=fp8 2>&1 | grep '"step"'
