"""Op dispatch: hand-written gfx950 HIP kernels on GPU, torch reference on CPU.

Policy (deliberate, per the framework's native-code contract):
- Tensors on a HIP device REQUIRE the in-tree native extension `_pa_hip`.
  If it is not importable, ops raise instead of silently falling back to
  eager PyTorch — a GPU run must execute our CDNA4 kernels or fail loudly.
  (Escape hatch for debugging only: PA_ALLOW_EAGER=1.)
- CPU tensors use ops.reference — the same functions that serve as the fp32
  ground truth in the kernel numerics tests.
- Shapes a kernel doesn't cover dispatch to reference WITH a one-time
  warning, so coverage gaps are visible in logs, not silent.

Plain GEMMs (nn.Linear / conv) go through PyTorch-ROCm's hipBLASLt/MIOpen
paths — library GEMMs are the sanctioned exception; the fused/bandwidth
ops here are the hand-written surface.
"""
from __future__ import annotations

import glob
import importlib.util
import logging
import math
import os
from typing import Optional

import torch

from . import reference

log = logging.getLogger("parallelanything.ops")

_EXT = None
_EXT_TRIED = False
_ALLOW_EAGER = os.environ.get("PA_ALLOW_EAGER", "0") == "1"
_WARNED: set = set()


def _load_ext():
    global _EXT, _EXT_TRIED
    if _EXT_TRIED:
        return _EXT
    _EXT_TRIED = True
    pkg_dir = os.path.dirname(__file__)
    cands = glob.glob(os.path.join(pkg_dir, "_pa_hip*.so"))
    if not cands:
        return None
    try:
        import torch  # noqa: F401  (must be imported before the ext)

        spec = importlib.util.spec_from_file_location("_pa_hip", cands[0])
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        _EXT = mod
        log.info("loaded HIP extension %s", cands[0])
    except Exception as err:  # noqa: BLE001
        log.error("failed to load HIP extension: %r", err)
        _EXT = None
    return _EXT


def hip_ext():
    return _load_ext()


def hip_available(op: Optional[str] = None) -> bool:
    ext = _load_ext()
    if ext is None:
        return False
    return op is None or hasattr(ext, op)


def _require_ext(op: str):
    ext = _load_ext()
    if ext is not None and hasattr(ext, op):
        return ext
    if _ALLOW_EAGER:
        if op not in _WARNED:
            _WARNED.add(op)
            log.warning("PA_ALLOW_EAGER: %s falling back to eager on GPU", op)
        return None
    raise RuntimeError(
        f"HIP extension for op '{op}' is not available on this GPU build — "
        "build it in-tree with `python -m comfyui_parallelanything_amd.ops.build` "
        "(gfx950). Refusing silent eager fallback."
    )


def _unsupported(op: str, why: str):
    key = (op, why)
    if key not in _WARNED:
        _WARNED.add(key)
        log.warning("op %s: shape/dtype not covered by HIP kernel (%s); "
                    "using reference path", op, why)


# ---------------------------------------------------------------------------
# Public ops
# ---------------------------------------------------------------------------

def rms_norm(x: torch.Tensor, weight=None, eps: float = 1e-6) -> torch.Tensor:
    if x.is_cuda:
        ext = _require_ext("rms_norm")
        if ext is not None:
            return ext.rms_norm(x.contiguous(),
                                weight.contiguous() if weight is not None else None,
                                eps)
    return reference.rms_norm(x, weight, eps)


def layer_norm_mod(x, scale, shift, eps: float = 1e-6) -> torch.Tensor:
    if x.is_cuda:
        ext = _require_ext("layer_norm_mod")
        if ext is not None:
            if scale.dim() == x.dim() - 1:
                return ext.layer_norm_mod(
                    x.contiguous(), scale.contiguous(), shift.contiguous(), eps
                )
            _unsupported("layer_norm_mod", "per-token modulation")
    return reference.layer_norm_mod(x, scale, shift, eps)


def gate_residual(residual, gate, x) -> torch.Tensor:
    if x.is_cuda:
        ext = _require_ext("gate_residual")
        if ext is not None and gate.dim() == x.dim() - 1:
            return ext.gate_residual(residual.contiguous(), gate.contiguous(),
                                     x.contiguous())
    return reference.gate_residual(residual, gate, x)


def group_norm_silu(x, num_groups: int, weight=None, bias=None,
                    eps: float = 1e-6) -> torch.Tensor:
    if x.is_cuda:
        ext = _require_ext("group_norm_silu")
        if ext is not None:
            if x.dim() == 4 and (x.shape[1] % num_groups) == 0:
                return ext.group_norm_silu(
                    x.contiguous(), int(num_groups),
                    weight.contiguous() if weight is not None else None,
                    bias.contiguous() if bias is not None else None,
                    eps,
                )
            _unsupported("group_norm_silu", f"dim={x.dim()}")
    return reference.group_norm_silu(x, num_groups, weight, bias, eps)


def rope_apply(x, cs) -> torch.Tensor:
    if x.is_cuda:
        ext = _require_ext("rope_apply")
        if ext is not None:
            return ext.rope_apply(x.contiguous(), cs.contiguous())
    return reference.rope_apply(x, cs)


def rope_freqs(positions, dim: int, theta: float = 10000.0) -> torch.Tensor:
    # Table precompute: once per replica per resolution, on the owning GPU
    # (the reference scrubbed these caches off replicas instead —
    # clear_flux_caches, any_device_parallel.py:166-195).
    return reference.rope_freqs(positions, dim, theta)


def attention(q, k, v, scale: Optional[float] = None) -> torch.Tensor:
    """Flash-style fused attention, layout [B, H, S, D]."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if q.is_cuda:
        ext = _require_ext("attn_fwd")
        if ext is not None:
            D = q.shape[-1]
            if q.dtype == torch.bfloat16 and D in (64, 128):
                return ext.attn_fwd(q.contiguous(), k.contiguous(),
                                    v.contiguous(), float(scale))
            _unsupported("attn_fwd", f"dtype={q.dtype}, D={D}")
            return reference.attention(q, k, v, scale)
    return reference.attention(q, k, v, scale)


def attention_bshd(q, k, v, scale: Optional[float] = None) -> torch.Tensor:
    """Fused attention on [B, S, H, D] (strided views allowed — e.g. slices
    of a fused qkv projection; no transpose/contiguous copies on GPU).

    Head dims that are not 64/128 but fit under them (SD1.5's 40/80) are
    zero-padded: padding q/k leaves every softmax score unchanged, padded v
    columns are sliced off the output.

    Cost statement: the pad wastes attention FLOPs
    proportional to Dp/D — 60% at D=40->64, 60% at D=80->128 — plus one
    pad copy of q/k/v per call. This is acceptable because SD1.5 is the
    PLUMBING config (BASELINE config 1 runs it on cpu,cpu); no SD-class
    GPU performance is claimed anywhere. A native D=48/96 tile would need
    fractional per-thread stage vectors (KVBLK*D/8 not divisible by 512
    threads) — not worth the schedule split while no headline depends on
    these shapes.
    """
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if q.is_cuda:
        ext = _require_ext("attn_fwd_bshd")
        if ext is not None:
            D = q.shape[-1]
            if q.dtype == torch.bfloat16 and D in (64, 128):
                return ext.attn_fwd_bshd(q, k, v, float(scale))
            if q.dtype == torch.bfloat16 and D < 128:
                Dp = 64 if D <= 64 else 128
                pad = (0, Dp - D)
                out = ext.attn_fwd_bshd(
                    torch.nn.functional.pad(q, pad),
                    torch.nn.functional.pad(k, pad),
                    torch.nn.functional.pad(v, pad),
                    float(scale),
                )
                return out[..., :D]
            _unsupported("attn_fwd_bshd", f"dtype={q.dtype}, D={D}")
    out = reference.attention(
        q.permute(0, 2, 1, 3), k.permute(0, 2, 1, 3), v.permute(0, 2, 1, 3),
        scale,
    )
    return out.permute(0, 2, 1, 3)


def attention_bshd_split(q, k, v, split: int, scale: Optional[float] = None):
    """attention_bshd with per-stream outputs: rows [:split] and [split:]
    land in two contiguous tensors (feeds dual-stream projections with no
    reshape copies)."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if q.is_cuda:
        if os.environ.get("PA_ATTN_V3") == "1":
            # the v3 debug kernel has no split-output epilogue: compose
            # via the plain kernel + slices so the escape hatch covers
            # the whole op surface
            out = attention_bshd(q, k, v, scale)
            return out[:, :split].contiguous(), out[:, split:].contiguous()
        ext = _require_ext("attn_fwd_bshd_split")
        if ext is not None and q.dtype == torch.bfloat16 \
                and q.shape[-1] in (64, 128):
            a, b = ext.attn_fwd_bshd_split(q, k, v, float(scale), int(split))
            return a, b
        _unsupported("attn_fwd_bshd_split",
                     f"dtype={q.dtype}, D={q.shape[-1]}")
    out = attention_bshd(q, k, v, scale)
    return out[:, :split].contiguous(), out[:, split:].contiguous()


def qk_norm_rope_(q, k, wq, wk, cs, eps: float = 1e-6):
    """In-place fused qk RMSNorm + RoPE on [B, S, H, D] views.

    Replaces four bandwidth passes (rms q, rms k, rope q, rope k) with one;
    per-replica cs tables live on the owning GPU (see models.layers).
    """
    if q.is_cuda:
        ext = _require_ext("qk_norm_rope_")
        if ext is not None and q.dtype == torch.bfloat16 and q.shape[-1] <= 128:
            ext.qk_norm_rope_(q, k, wq, wk, cs, eps)
            return q, k
        _unsupported("qk_norm_rope_", f"dtype={q.dtype}, D={q.shape[-1]}")
    qn = reference.rms_norm(q, wq, eps).permute(0, 2, 1, 3)
    kn = reference.rms_norm(k, wk, eps).permute(0, 2, 1, 3)
    q.copy_(reference.rope_apply(qn, cs).permute(0, 2, 1, 3))
    k.copy_(reference.rope_apply(kn, cs).permute(0, 2, 1, 3))
    return q, k


def pack_joint_qkv(txt_qkv, img_qkv, wq_t, wk_t, wq_i, wk_i, cs,
                   eps: float = 1e-6):
    """Dual-stream joint qkv pack: per-stream qk RMSNorm + RoPE + contiguous
    [B, T+Si, H, D] q/k/v in one pass (txt_qkv/img_qkv: [B,S,3,H,D] views)."""
    if txt_qkv.is_cuda:
        ext = _require_ext("pack_joint_qkv")
        if ext is not None and txt_qkv.dtype == torch.bfloat16                 and txt_qkv.shape[-1] <= 128:
            return ext.pack_joint_qkv(txt_qkv, img_qkv, wq_t, wk_t, wq_i, wk_i,
                                      cs, eps)
        _unsupported("pack_joint_qkv",
                     f"dtype={txt_qkv.dtype}, D={txt_qkv.shape[-1]}")
    T = txt_qkv.shape[1]
    tq, tk, tv = txt_qkv.unbind(2)
    iq, ik, iv = img_qkv.unbind(2)
    qk_norm_rope_(tq, tk, wq_t, wk_t, cs[:T], eps)
    qk_norm_rope_(iq, ik, wq_i, wk_i, cs[T:], eps)
    q = torch.cat([tq, iq], dim=1)
    k = torch.cat([tk, ik], dim=1)
    v = torch.cat([tv, iv], dim=1)
    return q, k, v


def quant_fp8(x: torch.Tensor, scale: torch.Tensor,
              amax_buf: torch.Tensor,
              scale_used: torch.Tensor | None = None) -> torch.Tensor:
    """Fused bf16 -> e4m3fn cast with running-amax update (delayed scaling).

    The fused epilogue overwrites ``scale[0]`` with the NEXT call's scale;
    pass a separate ``scale_used`` buffer to receive a snapshot of the scale
    this call actually quantized with (what _scaled_mm must dequantize by).
    GPU-only; used by the fp8 serving mode."""
    ext = _require_ext("quant_fp8")
    return ext.quant_fp8(x, scale, amax_buf,
                         scale if scale_used is None else scale_used)


def gelu_fp8(x, scale, amax_buf, scale_used) -> torch.Tensor:
    """tanh-GELU emitting e4m3fn with quant_fp8's delayed-scaling contract
    (fp8 serving mode: the MLP-down GEMM consumes the output directly,
    no standalone quant pass). GPU-only."""
    ext = _require_ext("gelu_fp8")
    return ext.gelu_fp8(x, scale, amax_buf, scale_used)


def gelu_tanh(x: torch.Tensor) -> torch.Tensor:
    if x.is_cuda:
        ext = _require_ext("gelu_tanh")
        if ext is not None:
            return ext.gelu_tanh(x)
    return torch.nn.functional.gelu(x, approximate="tanh")


def timestep_embedding(t, dim: int, max_period: float = 10000.0,
                       time_factor: float = 1000.0) -> torch.Tensor:
    if t.is_cuda:
        ext = _require_ext("timestep_embedding")
        if ext is not None:
            return ext.timestep_embedding(t.contiguous().float(), int(dim),
                                          float(max_period), float(time_factor))
    return reference.timestep_embedding(t, dim, max_period, time_factor)


def layer_norm_mod_fp8(x, scale, shift, qscale, amax_buf, scale_used,
                       eps: float = 1e-6) -> torch.Tensor:
    """AdaLN-modulated LayerNorm emitting e4m3fn directly (fp8 serving
    mode): the bf16->fp8 cast fuses into the normalize pass, replacing
    layer_norm_mod + quant_fp8 (two HBM round-trips -> one). Delayed
    scaling matches quant_fp8: quantizes with qscale[0], updates the
    running amax, writes the next scale, snapshots the used scale into
    scale_used. GPU-only."""
    ext = _require_ext("layer_norm_mod_fp8")
    return ext.layer_norm_mod_fp8(x, scale, shift, qscale, amax_buf,
                                  scale_used, float(eps))


def timestep_embed_mlp(t, w1, b1, max_period: float = 10000.0,
                       time_factor: float = 1000.0) -> torch.Tensor:
    """Fused sinusoidal embedding + first Linear + SiLU: [B] -> [B, H].

    SURVEY §2b timestep-MLP fusion (the in_layer half; the out Linear is a
    GEMM and stays on hipBLASLt). GPU-only — callers fall back to the
    composed path on CPU (MLPEmbedder.forward_timestep)."""
    ext = _require_ext("timestep_embed_mlp")
    return ext.timestep_embed_mlp(t, w1, b1, float(max_period),
                                  float(time_factor))
