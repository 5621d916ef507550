import time, torch, sys
sys.path.insert(0, ".")
import resilient_llm_amd.ops as ops

def t(fn, n=200):
    for _ in range(20): fn()
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    return (time.monotonic() - t0) / n * 1e6

for B in (64, 256):
    x = torch.randn(B, 4096, dtype=torch.bfloat16, device="cuda")
    res = torch.randn_like(x)
    w = torch.randn(4096, dtype=torch.bfloat16, device="cuda")
    us = t(lambda: ops.rmsnorm_residual_(x, res, w, 1e-5))
    gb = B * 4096 * 2 * 4 / 1e9   # x+res reads, res+y writes
    print(f"rmsnorm  B={B}: {us:6.2f} us  {gb/us*1e6/1e3:5.2f} TB/s")
    gu = torch.randn(B, 2*14336, dtype=torch.bfloat16, device="cuda")
    us = t(lambda: ops.silu_mul(gu))
    gb = B * 14336 * 2 * 3 / 1e9
    print(f"silu_mul B={B}: {us:6.2f} us  {gb/us*1e6/1e3:5.2f} TB/s")
