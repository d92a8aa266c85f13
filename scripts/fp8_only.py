"""Drive the three fp8-emitting kernels for PMC collection."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from comfyui_parallelanything_amd import ops

x_g = torch.randn(8 * 4608, 12288, device="cuda", dtype=torch.bfloat16)
x_l = torch.randn(8, 4608, 3072, device="cuda", dtype=torch.bfloat16)
sc = torch.randn(8, 3072, device="cuda", dtype=torch.bfloat16) * 0.1
sh = torch.randn(8, 3072, device="cuda", dtype=torch.bfloat16) * 0.1
s1, a1, u1 = (torch.tensor([0.01], device="cuda"), torch.zeros(2, device="cuda"),
              torch.zeros(1, device="cuda"))
s2, a2, u2 = (torch.tensor([0.01], device="cuda"), torch.zeros(2, device="cuda"),
              torch.zeros(1, device="cuda"))
s3, a3, u3 = (torch.tensor([0.01], device="cuda"), torch.zeros(2, device="cuda"),
              torch.zeros(1, device="cuda"))
for _ in range(8):
    ops.gelu_fp8(x_g, s1, a1, u1)
    ops.quant_fp8(x_l.reshape(-1, 3072), s2, a2, scale_used=u2)
    ops.layer_norm_mod_fp8(x_l, sc, sh, s3, a3, u3)
torch.cuda.synchronize()
print("done")
