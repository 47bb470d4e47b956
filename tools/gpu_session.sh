set -x
python -m pytest tests/test_gpu_kernels.py -m gpu -q 2>&1 | tail -2
python - <<'PY'
import sys, torch; sys.path.insert(0, '.')
from lstm_tensorspark_amd.ops import extension
ext = extension.get_extension()
g = torch.Generator(device="cuda").manual_seed(0)
B, H, T = 4096, 1024, 32
pre = torch.randn(T, B, 4*H, generator=g, device="cuda").bfloat16()
w_h = (torch.randn(H, 4*H, generator=g, device="cuda") / H**0.5).bfloat16()
h0 = torch.randn(B, H, generator=g, device="cuda").bfloat16()
c0 = torch.randn(B, H, generator=g, device="cuda")
def time_seq(mode):
    ext.set_split_fwd(mode)
    for _ in range(3): ext.lstm_seq_fwd(pre, w_h, h0, c0, False)
    torch.cuda.synchronize()
    s, e = torch.cuda.Event(enable_timing=True), torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(10): out = ext.lstm_seq_fwd(pre, w_h, h0, c0, False)
    e.record(); torch.cuda.synchronize()
    return s.elapsed_time(e)/10, out
t_f, out_f = time_seq(0)
t_s, out_s = time_seq(1)
err = max((a.float()-b.float()).abs().max().item() for a,b in zip(out_f, out_s))
print(f"seq_fwd fused={1e3*t_f/T:.1f}us/step split={1e3*t_s/T:.1f}us/step err={err:.4f}")
ext.set_split_fwd(-1)
PY
timeout 300 python bench.py --gpus 1 --steps 30 --warmup 5 2>&1 | tail -1
