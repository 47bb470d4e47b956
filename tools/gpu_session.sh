python - <<'PY'
import torch, time
torch.cuda.init()
R, H, H4 = 2097152, 1024, 4096   # T*B at headline, dw shape
dg = torch.randn(R, H4, device="cuda").bfloat16() * 0.01
hp = torch.randn(R, H, device="cuda").bfloat16()
def t(fn, n=6):
    for _ in range(2): fn()
    torch.cuda.synchronize(); s=torch.cuda.Event(enable_timing=True); e=torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(n): fn()
    e.record(); torch.cuda.synchronize(); return s.elapsed_time(e)/n
ms = t(lambda: hp.t() @ dg)
fl = 2.0*R*H*H4
print(f"bf16 dw GEMM: {ms:.2f} ms ({fl/ms/1e9:.0f} TF/s)")
hp8t = (hp.t().contiguous() * 448).to(torch.float8_e4m3fn)
dg8cm = (dg * 100).to(torch.float8_e4m3fn).t().contiguous().t()
sa = torch.full((), 1/448., device="cuda"); sb = torch.full((), 1/100., device="cuda")
ms8 = t(lambda: torch._scaled_mm(hp8t, dg8cm, scale_a=sa, scale_b=sb, out_dtype=torch.bfloat16))
print(f"fp8 dw GEMM: {ms8:.2f} ms ({fl/ms8/1e9:.0f} TF/s)")
# dx shape: (2M,4096)x(4096,1024)
wx = torch.randn(H, H4, device="cuda").bfloat16()
msx = t(lambda: dg @ wx.t())
print(f"bf16 dx GEMM: {msx:.2f} ms ({2.0*R*H4*H/msx/1e9:.0f} TF/s)")
dg8 = (dg*100).to(torch.float8_e4m3fn)
wx8t = (wx*100).to(torch.float8_e4m3fn).t()
msx8 = t(lambda: torch._scaled_mm(dg8, wx8t, scale_a=sb, scale_b=sb, out_dtype=torch.bfloat16))
print(f"fp8 dx GEMM: {msx8:.2f} ms ({2.0*R*H4*H/msx8/1e9:.0f} TF/s)")
PY
