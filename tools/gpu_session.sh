set -x
python -m pytest tests -m gpu -q 2>&1 | tail -1
timeout 400 python bench.py --gpus 1 --steps 6 --warmup 2 --preset bidir-4l 2>&1 | tail -1
timeout 400 python bench.py --gpus 1 --steps 6 --warmup 2 --preset lm-50k 2>&1 | tail -1
timeout 400 python bench.py --gpus 1 --steps 15 --warmup 3 2>&1 | tail -1
python - <<'PY'
import sys, torch; sys.path.insert(0, '.')
from lstm_tensorspark_amd.ops import extension
ext = extension.get_extension()
g = torch.Generator(device="cuda").manual_seed(0)
for B in (256, 512, 1024):
    H, T = 1024, 16
    pre = torch.randn(T, B, 4*H, generator=g, device="cuda").bfloat16()
    w_h = (torch.randn(H, 4*H, generator=g, device="cuda") / H**0.5).bfloat16()
    h0 = torch.randn(B, H, generator=g, device="cuda").bfloat16()
    c0 = torch.randn(B, H, generator=g, device="cuda")
    for mode in (0, 1):
        ext.set_split_fwd(mode)
        for _ in range(3): ext.lstm_seq_fwd(pre, w_h, h0, c0, False)
        torch.cuda.synchronize()
        s, e = torch.cuda.Event(enable_timing=True), torch.cuda.Event(enable_timing=True)
        s.record()
        for _ in range(10): ext.lstm_seq_fwd(pre, w_h, h0, c0, False)
        e.record(); torch.cuda.synchronize()
        print(f"B={B} split={mode}: {1e3*s.elapsed_time(e)/10/T:.1f} us/step")
ext.set_split_fwd(-1)
PY
