set -x
python - <<'PY'
import sys, torch, time; sys.path.insert(0, '.')
from lstm_tensorspark_amd.config import get_preset
from lstm_tensorspark_amd.trainer import Trainer
from lstm_tensorspark_amd.ops.lstm_layer import enable_side_dw
cfg = get_preset("mi355x-1gpu"); cfg.train.log_interval = 0; cfg.train.ckpt_dir = ""
t = Trainer(cfg, device="cuda:0")
g = torch.Generator(device="cuda"); g.manual_seed(1)
B, T = cfg.train.batch_size, cfg.train.seq_len
x = torch.randn(T, B, 1024, generator=g, device="cuda").bfloat16()
y = torch.randint(0, 128, (T, B), generator=g, device="cuda")
for side in (True, False, True):
    enable_side_dw(side)
    for _ in range(2): t.train_step((x, y))
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(6): t.train_step((x, y))
    torch.cuda.synchronize()
    print(f"side_dw={side}: {(time.perf_counter()-t0)/6*1e3:.1f} ms/step")
PY
