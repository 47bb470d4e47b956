set -x
python -m pytest tests -m gpu -q 2>&1 | tail -1
timeout 400 python bench.py --gpus 1 --steps 6 --warmup 2 --preset bidir-4l 2>&1 | tail -1
python - <<'PY'
import sys, torch, time; sys.path.insert(0, '.')
from lstm_tensorspark_amd.config import get_preset
from lstm_tensorspark_amd.trainer import Trainer
from lstm_tensorspark_amd.ops import extension
ext = extension.get_extension()
cfg = get_preset("bidir-4l"); cfg.train.log_interval = 0; cfg.train.ckpt_dir = ""
t = Trainer(cfg, device="cuda:0")
b = t.data.batch(0)
for mode, name in [(-1, "auto(fused@B1024)"), (1, "split")]:
    ext.set_split_fwd(mode)
    for _ in range(2): t.train_step(b)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(4): t.train_step(b)
    torch.cuda.synchronize()
    print(f"bidir split={name}: {(time.perf_counter()-t0)/4*1e3:.0f} ms/step")
ext.set_split_fwd(-1)
PY
