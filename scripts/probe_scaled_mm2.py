import torch
dev = "cuda:0"
torch.manual_seed(0)
M, N, K = 16, 32, 64
x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.05
sx = (x.abs().max().float() / 448.0)
sw = (w.abs().max().float() / 448.0)
x8 = (x.float() / sx).to(torch.float8_e4m3fn)
w8 = (w.float() / sw).to(torch.float8_e4m3fn)
ref = (x8.float() * sx) @ (w8.float() * sw).t()
for tag, kw in [
    ("0dim", dict(scale_a=sx, scale_b=sw)),
    ("1dim", dict(scale_a=sx.reshape(1), scale_b=sw.reshape(1))),
]:
    try:
        out = torch._scaled_mm(x8, w8.t(), out_dtype=torch.bfloat16, **kw)
        rel = (out.float() - ref).abs().max() / ref.abs().max()
        print(tag, "rel", rel.item())
    except Exception as e:
        print(tag, "FAIL", str(e)[:200])
# fnuz probe
try:
    x8z = (x.float() / sx).to(torch.float8_e4m3fnuz)
    w8z = (w.float() / sw).to(torch.float8_e4m3fnuz)
    out = torch._scaled_mm(x8z, w8z.t(), scale_a=sx, scale_b=sw,
                           out_dtype=torch.bfloat16)
    rel = (out.float() - ref).abs().max() / ref.abs().max()
    print("fnuz rel", rel.item())
except Exception as e:
    print("fnuz FAIL", str(e)[:200])
print("ref row0:", ref[0, :4].tolist())
out = torch._scaled_mm(x8, w8.t(), scale_a=sx, scale_b=sw, out_dtype=torch.bfloat16)
print("out row0:", out[0, :4].float().tolist())
