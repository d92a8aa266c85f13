"""FP8 (OCP e4m3fn) serving mode for gfx950.

MI355X runs fp8 MFMA at ~2x the bf16 rate (~5 PF dense). This module swaps
large nn.Linear layers for FP8Linear: weights quantized once to e4m3fn with
a per-tensor scale, activations quantized dynamically per call, matmul via
torch._scaled_mm (hipBLASLt fp8 path). Norms/attention stay bf16.

This is an OPT-IN mode (bench --dtype fp8; quantize_fp8(model)). The
flagship benchmark stays bf16 — reduced-precision numbers are reported
separately, never as the headline (bench contract).
"""
from __future__ import annotations

import logging

import torch
from torch import nn

log = logging.getLogger("parallelanything")

FP8 = torch.float8_e4m3fn
FP8_MAX = 448.0


def _supports_scaled_mm() -> bool:
    if not torch.cuda.is_available():
        return False
    try:
        a = torch.randn(16, 16, device="cuda").to(FP8)
        b = torch.randn(16, 16, device="cuda").to(FP8).t().contiguous().t()
        s = torch.ones((), device="cuda")
        torch._scaled_mm(a, b, scale_a=s, scale_b=s, out_dtype=torch.bfloat16)
        return True
    except Exception as err:  # noqa: BLE001
        log.warning("fp8 _scaled_mm unavailable: %r", err)
        return False


class FP8Linear(nn.Module):
    """Linear with e4m3fn weights + per-tensor scales (dynamic act quant)."""

    def __init__(self, weight_fp8: torch.Tensor, w_scale: torch.Tensor,
                 bias: torch.Tensor | None, out_dtype: torch.dtype):
        super().__init__()
        # weight stored [in, out] column-major-for-B as _scaled_mm wants
        self.register_buffer("weight_fp8", weight_fp8, persistent=True)
        self.register_buffer("w_scale", w_scale, persistent=True)
        if bias is not None:
            self.register_buffer("bias", bias, persistent=True)
        else:
            self.bias = None
        self.out_dtype = out_dtype
        self.in_features = weight_fp8.shape[1]
        self.out_features = weight_fp8.shape[0]
        # delayed activation scaling: running amax + next-call scale are
        # maintained ENTIRELY by the quant kernel (slot 1 of a_amax is its
        # block counter; see pa_ops.hip quant_fp8 fused epilogue) — zero
        # per-call host-launched scale math. The scale used is the
        # PREVIOUS call's. Buffers MUST live on the weight's device — the
        # module is swapped in place and never .to()-ed afterwards.
        self.register_buffer(
            "a_amax",
            torch.zeros(2, dtype=torch.float32, device=weight_fp8.device),
            persistent=False,
        )
        self.register_buffer(
            "x_scale",
            torch.full((1,), 1.0, dtype=torch.float32,
                       device=weight_fp8.device),
            persistent=False,
        )
        # Snapshot of the scale the quant kernel actually used this call:
        # its epilogue overwrites x_scale[0] with the NEXT call's scale, so
        # _scaled_mm must dequantize from this buffer, never x_scale itself
        # (same-buffer use multiplied the output by next_scale/used_scale).
        # A persistent separate buffer keeps all kernel pointers fixed
        # across calls, so hipGraph capture/replay stays valid.
        self.register_buffer(
            "x_scale_used",
            torch.full((1,), 1.0, dtype=torch.float32,
                       device=weight_fp8.device),
            persistent=False,
        )
        self._warm = False

    @classmethod
    def from_linear(cls, lin: nn.Linear) -> "FP8Linear":
        w = lin.weight.detach().float()  # [out, in]
        w_scale = (w.abs().amax() / FP8_MAX).clamp(min=1e-12)
        w8 = (w / w_scale).clamp(-FP8_MAX, FP8_MAX).to(FP8)
        # _scaled_mm wants B as [K, N] column-major: pass w8.t() (a view of
        # the row-major [out, in] weight IS column-major [in, out])
        bias = lin.bias.detach().clone() if lin.bias is not None else None
        return cls(w8, w_scale.to(w.device), bias, lin.weight.dtype)

    def _warm_from(self, amax0: torch.Tensor) -> None:
        self.a_amax[:1].copy_(amax0)
        self.x_scale.copy_((amax0 / FP8_MAX).clamp(min=1e-12))
        self._warm = True

    def mm_fp8(self, x8: torch.Tensor, scale_used: torch.Tensor) -> torch.Tensor:
        """fp8 GEMM on an already-quantized activation (+ its used scale)."""
        shape = x8.shape
        y = torch._scaled_mm(
            x8.reshape(-1, shape[-1]), self.weight_fp8.t(),
            scale_a=scale_used.reshape(()), scale_b=self.w_scale,
            bias=self.bias.to(self.out_dtype) if self.bias is not None else None,
            out_dtype=self.out_dtype,
        )
        return y.reshape(*shape[:-1], y.shape[-1])

    def ln_quant(self, x: torch.Tensor, mscale: torch.Tensor,
                 mshift: torch.Tensor):
        """AdaLN-modulated LayerNorm with the fp8 cast fused into the LN
        kernel (no standalone quant pass). Returns (x8 [B,S,D], scale_used).
        Callers may share x8/scale across several consumers of the same
        normalized activation (e.g. single-block qkv + mlp-up)."""
        from .. import ops

        if not self._warm:
            ln = ops.layer_norm_mod(x, mscale, mshift)
            self._warm_from(ln.abs().amax().reshape(1).float())
        x8 = ops.layer_norm_mod_fp8(x, mscale, mshift, self.x_scale,
                                    self.a_amax, self.x_scale_used)
        return x8, self.x_scale_used

    def gelu_quant(self, y: torch.Tensor):
        """tanh-GELU with the fp8 cast fused (this Linear consumes the
        GELU output): returns (x8, scale_used) for mm_fp8."""
        from .. import ops

        if not self._warm:
            g = torch.nn.functional.gelu(y.float(), approximate="tanh")
            self._warm_from(g.abs().amax().reshape(1).float())
        x8 = ops.gelu_fp8(y, self.x_scale, self.a_amax, self.x_scale_used)
        return x8, self.x_scale_used

    def forward_ln(self, x: torch.Tensor, mscale: torch.Tensor,
                   mshift: torch.Tensor) -> torch.Tensor:
        """layer_norm_mod -> this Linear, with the quant fused into the LN
        kernel when available (fp8 serving hot path)."""
        from .. import ops

        if (x.is_cuda and x.dtype == torch.bfloat16
                and ops.hip_available("layer_norm_mod_fp8")):
            x8, s_used = self.ln_quant(x, mscale, mshift)
            return self.mm_fp8(x8, s_used)
        return self.forward(ops.layer_norm_mod(x, mscale, mshift))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        from .. import ops

        if not x.is_cuda:
            raise RuntimeError(
                "FP8Linear is GPU-only (gfx950 _scaled_mm): fp8-quantized "
                "models cannot run on cpu chain devices — drop the cpu "
                "entries or skip quantize_fp8"
            )
        shape = x.shape
        x2 = x.reshape(-1, shape[-1])
        if (x.is_cuda and x.dtype == torch.bfloat16
                and ops.hip_available("quant_fp8")):
            if not self._warm:
                # first call: measure directly (still async, device-side)
                self._warm_from(x2.abs().amax().reshape(1).float())
            # the kernel quantizes with x_scale, updates the running amax,
            # writes the NEXT call's x_scale, and snapshots the scale it
            # used into x_scale_used — no host-side scale math
            x8 = ops.quant_fp8(x2, self.x_scale, self.a_amax,
                               scale_used=self.x_scale_used)
            x_scale = self.x_scale_used
        else:
            x_scale = (x2.abs().amax().float() / FP8_MAX).clamp(min=1e-12)
            x8 = (x2.float() / x_scale).clamp(-FP8_MAX, FP8_MAX).to(FP8)
        y = torch._scaled_mm(
            x8, self.weight_fp8.t(),
            scale_a=x_scale.reshape(()), scale_b=self.w_scale,
            bias=self.bias.to(self.out_dtype) if self.bias is not None else None,
            out_dtype=self.out_dtype,
        )
        return y.reshape(*shape[:-1], y.shape[-1])


def quantize_fp8(model: nn.Module, min_features: int = 1024) -> int:
    """Swap large Linears for FP8Linear in place; returns the swap count.

    Small projections (timestep/vector embedders, modulation heads with
    tiny batch) stay bf16 — fp8 pays off on the token-major GEMMs.
    """
    if not _supports_scaled_mm():
        raise RuntimeError(
            "fp8 mode needs torch._scaled_mm on a gfx950 device"
        )
    n = 0
    for parent in model.modules():
        # GELULinear parents ARE quantized: its forward detects the
        # swapped .lin and runs fp8 GEMM + the standalone gelu kernel
        # (measured faster than keeping the bf16 _addmm_activation fusion:
        # the bf16 GELU-up GEMMs were ~112 ms of the 454 ms fp8 step).
        for name, child in list(parent.named_children()):
            if isinstance(child, nn.Linear) and (
                child.in_features >= min_features
                and child.out_features >= min_features
            ):
                setattr(parent, name, FP8Linear.from_linear(child))
                n += 1
    log.info("fp8: quantized %d Linear layers", n)
    return n
