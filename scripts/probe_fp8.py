import torch
print("has scaled_mm:", hasattr(torch, "_scaled_mm"))
a = torch.randn(64, 4096, device="cuda", dtype=torch.bfloat16)
w = torch.randn(6144, 4096, device="cuda", dtype=torch.bfloat16)
af = a.to(torch.float8_e4m3fn)
wf = w.to(torch.float8_e4m3fn)
sa = torch.tensor(1.0, device="cuda")
sb = torch.tensor(1.0, device="cuda")
try:
    out = torch._scaled_mm(af, wf.t(), scale_a=sa, scale_b=sb, out_dtype=torch.bfloat16)
    print("scaled_mm tensorwise ok", out.shape, out.dtype)
except Exception as e:
    print("tensorwise failed:", e)
# rowwise scales
try:
    sa_r = torch.ones(64, 1, device="cuda")
    sb_r = torch.ones(1, 6144, device="cuda")
    out = torch._scaled_mm(af, wf.t(), scale_a=sa_r, scale_b=sb_r, out_dtype=torch.bfloat16)
    print("scaled_mm rowwise ok", out.shape)
except Exception as e:
    print("rowwise failed:", e)
# timing
import time
def bench(fn, iters=50):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t=time.monotonic()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.monotonic()-t)/iters*1e6
t_bf16 = bench(lambda: a @ w.t())
t_fp8 = bench(lambda: torch._scaled_mm(af, wf.t(), scale_a=sa, scale_b=sb, out_dtype=torch.bfloat16))
print(f"bf16 {t_bf16:.1f}us fp8 {t_fp8:.1f}us speedup {t_bf16/t_fp8:.2f}x")
# bigger: gate_up shape
w2 = torch.randn(28672, 4096, device="cuda", dtype=torch.bfloat16)
w2f = w2.to(torch.float8_e4m3fn)
t2_bf16 = bench(lambda: a @ w2.t())
t2_fp8 = bench(lambda: torch._scaled_mm(af, w2f.t(), scale_a=sa, scale_b=sb, out_dtype=torch.bfloat16))
print(f"gate_up bf16 {t2_bf16:.1f}us fp8 {t2_fp8:.1f}us speedup {t2_bf16/t2_fp8:.2f}x")
