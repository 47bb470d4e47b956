set -x
python - <<'PY'
import sys, torch; sys.path.insert(0, '.')
from lstm_tensorspark_amd.ops import extension
ext = extension.get_extension()
g = torch.Generator(device="cuda").manual_seed(0)
B, H = 16384, 1024
h_prev = torch.randn(B, H, generator=g, device="cuda").bfloat16()
w_h = (torch.randn(H, 4*H, generator=g, device="cuda") / H**0.5).bfloat16()
pre = torch.randn(B, 4*H, generator=g, device="cuda").bfloat16()
c_prev = torch.randn(B, H, generator=g, device="cuda")
for v in (2, 32, 33, 40):
    ext.set_fwd_variant(v)
    for _ in range(5): ext.lstm_rec_fwd_step(h_prev, w_h, pre, c_prev, False)
    torch.cuda.synchronize()
    s, e = torch.cuda.Event(enable_timing=True), torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(20): ext.lstm_rec_fwd_step(h_prev, w_h, pre, c_prev, False)
    e.record(); torch.cuda.synchronize()
    us = 1e3*s.elapsed_time(e)/20
    print(f"B=16384 variant={v}: {us:.0f} us ({2*B*4*H*H/us/1e6:.0f} TF/s)")
ext.set_fwd_variant(-1)
PY
timeout 420 python train.py --preset mi355x-1gpu --set train.steps=300 --set train.log_interval=100 --set train.batch_size=8192 2>&1 | grep '"step"\|Error\|nan' | tail -4
rocm-smi --showmeminfo vram 2>/dev/null | grep Used
