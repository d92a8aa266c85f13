import sys, torch
sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
from comfyui_parallelanything_amd import ops
q = torch.randn(8, 24, 4608, 128, device="cuda", dtype=torch.bfloat16)
k = torch.randn_like(q); v = torch.randn_like(q)
for _ in range(3): ops.attention(q, k, v)
torch.cuda.synchronize()
for _ in range(10): ops.attention(q, k, v)
torch.cuda.synchronize()
