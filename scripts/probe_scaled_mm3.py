import torch
import torch.nn.functional as F
dev = "cuda:0"
torch.manual_seed(0)
M, N, K = 64, 28672, 4096

x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.05
# rowwise: per-row activation scale, per-column (out-channel) weight scale
sa = (x.float().abs().amax(dim=1, keepdim=True) / 448.0)      # [M,1]
sb = (w.float().abs().amax(dim=1, keepdim=True) / 448.0)      # [N,1]
x8 = (x.float() / sa).to(torch.float8_e4m3fn)
w8 = (w.float() / sb).to(torch.float8_e4m3fn)
ref = F.linear(x, w).float()
try:
    out = torch._scaled_mm(x8, w8.t(), scale_a=sa, scale_b=sb.t(),
                           out_dtype=torch.bfloat16)
    rel = (out.float() - ref).abs().max() / ref.abs().max()
    print("rowwise rel", rel.item())
except Exception as e:
    print("rowwise FAIL", str(e)[:200])

def timeit(fn, reps=60, warm=10):
    for _ in range(warm): fn()
    torch.cuda.synchronize()
    s, e = torch.cuda.Event(True), torch.cuda.Event(True)
    s.record()
    for _ in range(reps): fn()
    e.record(); torch.cuda.synchronize()
    return s.elapsed_time(e) / reps * 1e3

# L3-cold speed: cycle weight copies
C = 34
w8s = [w8.clone() for _ in range(C)]
i = [0]
def frow():
    i[0] = (i[0] + 1) % C
    return torch._scaled_mm(x8, w8s[i[0]].t(), scale_a=sa, scale_b=sb.t(),
                            out_dtype=torch.bfloat16)
sa0 = sa.amax().reshape(())
sb0 = sb.amax().reshape(())
def ftens():
    i[0] = (i[0] + 1) % C
    return torch._scaled_mm(x8, w8s[i[0]].t(), scale_a=sa0, scale_b=sb0,
                            out_dtype=torch.bfloat16)
try:
    print("rowwise us", timeit(frow))
except Exception as e:
    print("rowwise time FAIL", str(e)[:150])
print("tensorwise us", timeit(ftens))
