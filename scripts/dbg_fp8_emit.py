import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from resilient_llm_amd import ops
from resilient_llm_amd.ops import ref
torch.manual_seed(1)
dev = "cuda:0"
x = torch.randn(13, 1024, device=dev, dtype=torch.bfloat16)
res = torch.randn_like(x)
w = torch.randn(1024, device=dev, dtype=torch.bfloat16)
res_ref = res.clone()
q, s = ops.rmsnorm_residual_fp8(x, res, w)
qr, sr = ref.rmsnorm_residual_fp8(x.cpu(), res_ref.cpu(), w.cpu())
print("scales gpu:", s.cpu().tolist())
print("scales ref:", sr.tolist())
d = (q.float() * s.reshape(-1,1)) - (qr.float().to(dev) * sr.to(dev).reshape(-1,1))
print("per-row max err:", d.abs().amax(dim=1).tolist())
# also bf16 norm comparison
y = ops.rmsnorm_residual_(x.clone(), res.clone(), w)
