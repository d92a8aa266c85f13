"""In-process engine tests on a real HIP device (1 GPU is enough).

Covers the ComfyUI-node execution mode: lead-only on GPU, the mixed
cuda+cpu hybrid chain the reference supports (README.md:133-134 of the
reference), and pipeline mode with a GPU lead."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from comfyui_parallelanything_amd.models.registry import (
    flux_inputs,
    make_flux,
    make_sd15,
    sd15_inputs,
)
from comfyui_parallelanything_amd.parallel.chain import DeviceChain, make_entry
from comfyui_parallelanything_amd.parallel.cleanup import cleanup_parallel_model
from comfyui_parallelanything_amd.parallel.engine import (
    ParallelEngine,
    install_parallel_forward,
)
from comfyui_parallelanything_amd.parallel.pipeline import configure_pipeline


def chain(*entries):
    return DeviceChain.from_list([make_entry(d, p) for d, p in entries])


def test_lead_only_gpu():
    m = make_flux(dev="cuda:0", dtype=torch.bfloat16, tiny=True)
    eng = ParallelEngine(chain(("cuda:0", 100)), auto_vram_balance=False)
    eng.setup(m)
    x, t, c, kw = flux_inputs(2, dev="cuda:0", dtype=torch.bfloat16, tiny=True)
    out = eng.forward(x, t, context=c, **kw)
    assert out.shape == x.shape and out.is_cuda
    assert torch.isfinite(out.float()).all()


def test_hybrid_cuda_cpu_dp_matches_single():
    """[cuda:0 (50%), cpu (50%)] split == single-device output (fp32 tiny;
    the cpu replica runs the reference ops, the GPU replica the HIP ops —
    tolerance covers the kernel-vs-reference numerics)."""
    m = make_sd15(dev="cuda:0", dtype=torch.float32, tiny=True)
    x, t, c, kw = sd15_inputs(4, dev="cuda:0", tiny=True)
    ref = m(x, t, context=c, **kw)
    eng = ParallelEngine(
        chain(("cuda:0", 50), ("cpu", 50)), auto_vram_balance=False
    )
    eng.setup(m)
    out = eng.forward(x, t, context=c, **kw)
    assert out.device.type == "cuda"
    torch.testing.assert_close(out, ref, rtol=5e-3, atol=5e-3)
    eng.release()


def test_pipeline_mode_gpu_lead():
    m = make_flux(dev="cuda:0", dtype=torch.float32, tiny=True)
    x, t, c, kw = flux_inputs(1, dev="cuda:0", dtype=torch.float32, tiny=True)
    ref = m(x, t, context=c, **kw)
    eng = ParallelEngine(
        chain(("cuda:0", 50), ("cpu", 50)), auto_vram_balance=False
    )
    eng.setup(m, force_copy_lead=True)
    configure_pipeline(eng)
    assert eng.pipeline is not None
    out = eng.forward(x, t, context=c, **kw)
    torch.testing.assert_close(out, ref, rtol=5e-3, atol=5e-3)
    eng.release()


def test_microbatched_pipeline_gpu_lead():
    """batch 2 < 3 devices -> pipeline with 2 concurrent micro-batches on a
    [cuda, cuda, cpu] hybrid chain; matches the single-device forward."""
    m = make_flux(dev="cuda:0", dtype=torch.float32, tiny=True)
    x, t, c, kw = flux_inputs(2, dev="cuda:0", dtype=torch.float32, tiny=True)
    ref = m(x, t, context=c, **kw)
    eng = ParallelEngine(
        chain(("cuda:0", 40), ("cuda:0", 30), ("cpu", 30)),
        auto_vram_balance=False,
    )
    eng.setup(m, force_copy_lead=True)
    configure_pipeline(eng, microbatches=2)
    assert eng.pipeline is not None
    out = eng.forward(x, t, context=c, **kw)
    torch.testing.assert_close(out, ref, rtol=5e-3, atol=5e-3)
    eng.release()


def test_install_forward_cleanup_gpu():
    m = make_flux(dev="cuda:0", dtype=torch.bfloat16, tiny=True)
    eng = ParallelEngine(chain(("cuda:0", 100)), auto_vram_balance=False)
    eng.setup(m)
    install_parallel_forward(m, eng)
    x, t, c, kw = flux_inputs(2, dev="cuda:0", dtype=torch.bfloat16, tiny=True)
    out = m(x, t, context=c, **kw)
    assert out.shape == x.shape
    cleanup_parallel_model(m)
    assert not getattr(m, "_true_parallel_active", False)


def test_hip_graph_engine_matches_eager():
    """use_hip_graphs: first two calls are eager warmup, third captures,
    later calls replay — all must match the eager engine bit-for-bit-ish
    (same kernels, same order -> tight tolerance)."""
    m = make_flux(dev="cuda:0", dtype=torch.bfloat16, tiny=True)
    x, t, c, kw = flux_inputs(2, dev="cuda:0", dtype=torch.bfloat16, tiny=True)
    eager = ParallelEngine(chain(("cuda:0", 100)), auto_vram_balance=False)
    eager.setup(m)
    ref = eager.forward(x, t, context=c, **kw).clone()

    eng = ParallelEngine(
        chain(("cuda:0", 100)), auto_vram_balance=False, use_hip_graphs=True
    )
    eng.setup(m)
    for i in range(5):
        out = eng.forward(x, t, context=c, **kw)
        torch.cuda.synchronize()
        torch.testing.assert_close(out, ref, rtol=2e-2, atol=2e-2), i
    assert eng.graphs._graphs, "no hipGraph was captured after warmup"
    # changed shape -> new signature, falls back to eager warmup, still right
    x3, t3, c3, kw3 = flux_inputs(
        3, dev="cuda:0", dtype=torch.bfloat16, tiny=True
    )
    out3 = eng.forward(x3, t3, context=c3, **kw3)
    assert out3.shape == x3.shape
    assert torch.isfinite(out3.float()).all()
    eng.release()
    assert not eng.graphs._graphs


def test_vram_balancer_reads_hbm():
    from comfyui_parallelanything_amd.parallel.balance import get_free_vram_mb

    free = get_free_vram_mb("cuda:0")
    assert free > 10_000, f"expected >10 GB free HBM, got {free} MB"


@pytest.mark.parametrize("name", ["sdxl", "zimage", "sd3", "wan"])
def test_model_families_gpu_tiny(name):
    from comfyui_parallelanything_amd.models.registry import MODELS

    make, inputs = MODELS[name]
    m = make(dev="cuda:0", dtype=torch.bfloat16, tiny=True)
    x, t, c, kw = inputs(2, dev="cuda:0", dtype=torch.bfloat16, tiny=True)
    with torch.no_grad():
        out = m(x, t, context=c, **kw)
    assert out.shape == x.shape
    assert torch.isfinite(out.float()).all()


def test_fp8_model_under_hip_graphs():
    """fp8 serving mode inside engine hipGraph capture: the quant/LN/GELU
    fp8 kernels keep FIXED buffer pointers (scale, amax, snapshot), so
    graph replay must keep updating the delayed-scaling state and stay
    numerically sane across steps."""
    from comfyui_parallelanything_amd.models.quant import (
        _supports_scaled_mm, quantize_fp8,
    )
    from comfyui_parallelanything_amd.models.registry import (
        flux_inputs, make_flux,
    )
    from comfyui_parallelanything_amd.parallel.chain import DeviceChain
    from comfyui_parallelanything_amd.parallel.engine import ParallelEngine

    if not _supports_scaled_mm():
        pytest.skip("no fp8 _scaled_mm")
    m = make_flux(dev="cuda", dtype=torch.bfloat16, tiny=True)
    quantize_fp8(m, min_features=32)
    x, t, c, kw = flux_inputs(2, dev="cuda", dtype=torch.bfloat16, tiny=True)
    with torch.no_grad():
        ref = m(x, t, context=c, **kw).clone()
    eng = ParallelEngine(
        DeviceChain(devices=("cuda:0",), weights=(1.0,)),
        auto_vram_balance=False, use_hip_graphs=True,
    )
    eng.setup(m)
    outs = []
    for _ in range(5):  # 2 warmups -> capture -> replays
        outs.append(eng.forward(x, t, context=c, **kw).clone())
    assert eng.graphs._graphs, "graph was not captured"
    for o in outs:
        assert torch.isfinite(o.float()).all()
        # delayed scaling drifts slightly (0.999 decay) but replays must
        # stay close to the eager fp8 result
        rel = (o.float() - ref.float()).norm() / ref.float().norm()
        assert rel < 0.05, f"fp8 graph replay drifted: {rel:.4f}"
    eng.release()


def test_serve_fp8_mode_gpu():
    """The FastAPI surface with --fp8: quantized engine behind /generate."""
    from comfyui_parallelanything_amd.models.quant import _supports_scaled_mm

    if not _supports_scaled_mm():
        pytest.skip("no fp8 _scaled_mm")
    from fastapi.testclient import TestClient

    from comfyui_parallelanything_amd.serve import create_app

    app = create_app("flux", devices=["cuda:0"], tiny=True, fp8=True)
    client = TestClient(app)
    hz = client.get("/healthz").json()
    assert hz["fp8"] is True
    r = client.post("/generate", json={"batch": 2, "steps": 2, "seed": 1})
    assert r.status_code == 200
    body = r.json()
    assert body["finite"] is True and body["images_per_s"] > 0
