import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import copy, torch
from comfyui_parallelanything_amd.models.registry import MODELS
from comfyui_parallelanything_amd.models.quant import quantize_fp8

print("== cross-backend (GPU bf16 HIP vs CPU fp32 reference) rel-l2 ==")
for name in ["flux", "sd15", "zimage", "sd3", "wan_i2v"]:
    make, inputs = MODELS[name]
    m_ref = make(dev="cpu", dtype=torch.float32, tiny=True)
    m_gpu = copy.deepcopy(m_ref).to("cuda", torch.bfloat16)
    x, t, c, kw = inputs(2, tiny=True, dtype=torch.float32)
    with torch.no_grad():
        ref = m_ref(x, t, context=c, **kw).float()
        out = m_gpu(x.cuda().bfloat16(), t.cuda(), context=c.cuda().bfloat16(),
                    **{k: (v.cuda().bfloat16() if isinstance(v, torch.Tensor) else v)
                       for k, v in kw.items()}).float().cpu()
    rel = ((out - ref).norm() / ref.norm()).item()
    corr = torch.corrcoef(torch.stack([out.flatten(), ref.flatten()]))[0, 1].item()
    print(f"  {name:8s} rel {rel:.4f}  corr {corr:.5f}")

print("== fp8 serving mode vs bf16 (flux-tiny, settled scales) ==")
make, inputs = MODELS["flux"]
m = make(dev="cuda", dtype=torch.bfloat16, tiny=True)
x, t, c, kw = inputs(2, dev="cuda", dtype=torch.bfloat16, tiny=True)
with torch.no_grad():
    ref = m(x, t, context=c, **kw).float().clone()
    quantize_fp8(m, min_features=32)
    for _ in range(3):
        out = m(x, t, context=c, **kw).float()
rel = ((out - ref).norm() / ref.norm()).item()
print(f"  fp8 rel {rel:.4f}")
