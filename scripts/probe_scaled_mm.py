import torch
import torch.nn.functional as F
import time

dev = "cuda:0"
torch.manual_seed(0)
SHAPES = [("qkv", 6144, 4096), ("o", 4096, 4096),
          ("gate_up", 28672, 4096), ("down", 4096, 14336)]
M = 64
COPIES = 34

def timeit(fn, reps=60, warm=10):
    for _ in range(warm): fn()
    torch.cuda.synchronize()
    s, e = torch.cuda.Event(True), torch.cuda.Event(True)
    s.record()
    for _ in range(reps): fn()
    e.record(); torch.cuda.synchronize()
    return s.elapsed_time(e) / reps * 1e3

for name, N, K in SHAPES:
    x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    ws = [torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.02
          for _ in range(COPIES)]
    w8s = []
    for w in ws:
        scale = w.abs().max().float() / 448.0
        w8s.append(((w.float() / scale).to(torch.float8_e4m3fn), scale))
    xs = x.abs().max().float() / 448.0
    x8 = (x.float() / xs).to(torch.float8_e4m3fn)
    sx = xs.reshape(1)
    i = [0]
    def bf16():
        i[0] = (i[0] + 1) % COPIES
        return F.linear(x, ws[i[0]])
    def fp8():
        i[0] = (i[0] + 1) % COPIES
        w8, swc = w8s[i[0]]
        return torch._scaled_mm(x8, w8.t(), scale_a=sx,
                                scale_b=swc.reshape(1),
                                out_dtype=torch.bfloat16)
    try:
        out8 = fp8()
        ref = bf16()
        rel = (out8.float() - ref.float()).abs().max() / ref.float().abs().max()
        t16 = timeit(bf16)
        t8 = timeit(fp8)
        print(f"{name:8s} bf16 {t16:7.1f} us  fp8 {t8:7.1f} us  "
              f"speedup {t16/t8:4.2f}x  relerr {rel:.3f}", flush=True)
    except Exception as ex:
        print(f"{name}: FAIL {ex}", flush=True)
