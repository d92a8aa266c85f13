"""Shared building blocks for the diffusion model zoo.

All hot math routes through the ops dispatch layer (ops/__init__.py): fused
attention, AdaLN-modulated LayerNorm, RMSNorm, RoPE, gated residuals run as
hand-written gfx950 HIP kernels on GPU and as the fp32-able torch reference
on CPU. Plain projections stay nn.Linear (hipBLASLt on ROCm).

The architectures mirror the model families the reference node is documented
to run (README.md:5 of the reference: Z_IMAGE, FLUX.1, WAN2.2, plus the
SD/SDXL UNets every ComfyUI install has) — the reference contains no model
code itself; shapes here follow the public architectures.
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Tuple

import torch
from torch import nn

from .. import ops


class FusedGELU(nn.Module):
    """tanh-GELU through the ops dispatch (vectorized gfx950 kernel)."""

    def forward(self, x):
        return ops.gelu_tanh(x)


class GELULinear(nn.Module):
    """Linear with the GELU fused into the GEMM epilogue (hipBLASLt
    _addmm_activation on GPU; linear+tanh-gelu on CPU)."""

    def __init__(self, in_features: int, out_features: int):
        super().__init__()
        self.lin = nn.Linear(in_features, out_features, bias=True)

    def forward(self, x):
        if not isinstance(self.lin, nn.Linear):
            # fp8-swapped (models/quant.py): _scaled_mm has no GELU
            # epilogue, so run the fp8 GEMM + the standalone gelu kernel —
            # measured faster than keeping this GEMM bf16-fused.
            y = self.lin(x)
            if y.is_cuda:
                from .. import ops

                return ops.gelu_tanh(y)
            return torch.nn.functional.gelu(y, approximate="tanh")
        if x.is_cuda:
            x2 = x.reshape(-1, x.shape[-1])
            out = torch._addmm_activation(
                self.lin.bias, x2, self.lin.weight.t(), use_gelu=True
            )
            return out.reshape(*x.shape[:-1], out.shape[-1])
        return torch.nn.functional.gelu(self.lin(x), approximate="tanh")

    def forward_ln(self, x, scale, shift):
        """AdaLN-modulated LayerNorm feeding this projection; in fp8 mode
        the quant fuses into the LN kernel (models/quant.py ln_quant)."""
        ln_fwd = getattr(self.lin, "forward_ln", None)
        if ln_fwd is not None:
            y = ln_fwd(x, scale, shift)
            if y.is_cuda:
                return ops.gelu_tanh(y)
            return torch.nn.functional.gelu(y, approximate="tanh")
        return self.forward(ops.layer_norm_mod(x, scale, shift))


class FusedMLP(nn.Module):
    """Linear -> GELU (epilogue-fused) -> Linear."""

    def __init__(self, dim_in: int, dim_mid: int, dim_out: int):
        super().__init__()
        self.up = GELULinear(dim_in, dim_mid)
        self.down = nn.Linear(dim_mid, dim_out, bias=True)

    def _down_fp8(self, y):
        """fp8 mode: GELU + e4m3 cast in ONE kernel feeding the down GEMM
        (no standalone quant pass); None when the fused path is off."""
        if (hasattr(self.down, "gelu_quant") and y.is_cuda
                and y.dtype == torch.bfloat16
                and ops.hip_available("gelu_fp8")):
            x8, s_used = self.down.gelu_quant(y)
            return self.down.mm_fp8(x8, s_used)
        return None

    def forward(self, x):
        up_lin = getattr(self.up, "lin", None)
        if hasattr(up_lin, "mm_fp8"):
            y = up_lin(x)  # fp8 GEMM (bf16 out), GELU NOT yet applied
            out = self._down_fp8(y)
            if out is not None:
                return out
            return self.down(ops.gelu_tanh(y) if y.is_cuda else
                             torch.nn.functional.gelu(y, approximate="tanh"))
        return self.down(self.up(x))

    def forward_ln(self, x, scale, shift):
        up_lin = getattr(self.up, "lin", None)
        ln_fwd = getattr(up_lin, "forward_ln", None)
        if ln_fwd is not None:
            y = ln_fwd(x, scale, shift)  # fused LN+quant -> fp8 GEMM
            out = self._down_fp8(y)
            if out is not None:
                return out
            return self.down(ops.gelu_tanh(y) if y.is_cuda else
                             torch.nn.functional.gelu(y, approximate="tanh"))
        return self.down(self.up.forward_ln(x, scale, shift))


def ln_mod_into(proj, x, scale, shift):
    """layer_norm_mod -> proj; fp8-aware projections (FP8Linear,
    GELULinear, FusedMLP) fuse the e4m3 cast INTO the LN kernel via their
    own forward_ln — one HBM pass instead of LN + standalone quant."""
    fwd = getattr(proj, "forward_ln", None)
    if fwd is not None:
        return fwd(x, scale, shift)
    return proj(ops.layer_norm_mod(x, scale, shift))


class MLPEmbedder(nn.Module):
    """2-layer SiLU MLP used for timestep / vector conditioning."""

    def __init__(self, in_dim: int, hidden_dim: int):
        super().__init__()
        self.in_layer = nn.Linear(in_dim, hidden_dim, bias=True)
        self.out_layer = nn.Linear(hidden_dim, hidden_dim, bias=True)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.out_layer(torch.nn.functional.silu(self.in_layer(x)))

    def forward_timestep(self, t: torch.Tensor,
                         max_period: float = 10000.0,
                         time_factor: float = 1000.0) -> torch.Tensor:
        """Timestep scalar [B] -> conditioning vector [B, hidden].

        GPU: fused sinusoid+in_layer+SiLU kernel (SURVEY §2b timestep-MLP
        fusion; one launch instead of three and the sinusoid stays fp32
        into the accumulate), then the out_layer GEMM. CPU / non-bf16:
        composed ops."""
        # in_layer may be FP8Linear-swapped (no .weight attr): compose then
        w = getattr(self.in_layer, "weight", None)
        if (w is not None and t.is_cuda and w.dtype == torch.bfloat16
                and w.shape[1] % 8 == 0
                and ops.hip_available("timestep_embed_mlp")):
            h = ops.timestep_embed_mlp(t, w, self.in_layer.bias,
                                       max_period, time_factor)
            return self.out_layer(h)
        emb = ops.timestep_embedding(t, self.in_layer.in_features,
                                     max_period, time_factor)
        return self.forward(
            emb.to(w.dtype if w is not None else torch.bfloat16)
        )


class RMSNorm(nn.Module):
    def __init__(self, dim: int):
        super().__init__()
        self.scale = nn.Parameter(torch.ones(dim))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.rms_norm(x, self.scale)


class QKNorm(nn.Module):
    """Per-head-dim RMSNorm on q and k before attention (FLUX-style)."""

    def __init__(self, head_dim: int):
        super().__init__()
        self.query_norm = RMSNorm(head_dim)
        self.key_norm = RMSNorm(head_dim)

    def forward(self, q: torch.Tensor, k: torch.Tensor):
        return self.query_norm(q), self.key_norm(k)


@dataclass
class ModOut:
    shift: torch.Tensor
    scale: torch.Tensor
    gate: torch.Tensor


class Modulation(nn.Module):
    """vec -> (shift, scale, gate) x {1 or 2} AdaLN parameter sets."""

    def __init__(self, dim: int, double: bool):
        super().__init__()
        self.multiplier = 6 if double else 3
        self.lin = nn.Linear(dim, self.multiplier * dim, bias=True)

    def forward(self, vec: torch.Tensor):
        out = self.lin(torch.nn.functional.silu(vec))
        parts = out.chunk(self.multiplier, dim=-1)
        first = ModOut(*parts[:3])
        second = ModOut(*parts[3:]) if self.multiplier == 6 else None
        return first, second


def rope_2d_table(
    h: int, w: int, axes_dim: Tuple[int, ...], theta: float = 10000.0,
    device=None, text_len: int = 0,
) -> torch.Tensor:
    """RoPE cos/sin table for text+image joint sequence.

    axes_dim per-axis head-dim shares (FLUX: (16, 56, 56) summing to 128):
    axis 0 encodes a scalar index (0 for image tokens, position for text),
    axes 1/2 encode row/column. Text tokens get position ids on axis 0 and
    zeros elsewhere (FLUX txt_ids convention). Returns [S_txt + h*w, D/2, 2]
    fp32, precomputed once per replica per resolution on the owning device.
    """
    dev = device if device is not None else "cpu"
    ys, xs = torch.meshgrid(
        torch.arange(h, device=dev), torch.arange(w, device=dev), indexing="ij"
    )
    img_ids = torch.stack(
        [torch.zeros_like(ys), ys, xs], dim=-1
    ).reshape(-1, 3)  # [h*w, 3]
    txt_ids = torch.zeros(text_len, 3, device=dev)
    if text_len:
        txt_ids[:, 0] = torch.arange(text_len, device=dev)
    ids = torch.cat([txt_ids, img_ids.float()], dim=0)  # [S, 3]
    tables = [
        ops.rope_freqs(ids[:, ax], axes_dim[ax], theta) for ax in range(len(axes_dim))
    ]  # each [S, axes_dim[ax]/2, 2]
    return torch.cat(tables, dim=1)  # [S, D/2, 2]


class JointAttention(nn.Module):
    """Attention core shared by MMDiT blocks (scale holder; the heavy
    lifting is ops.attention_bshd on fused-qkv views)."""

    def __init__(self, num_heads: int, head_dim: int):
        super().__init__()
        self.num_heads = num_heads
        self.head_dim = head_dim
        self.scale = 1.0 / math.sqrt(head_dim)


class DoubleStreamBlock(nn.Module):
    """FLUX-class dual-stream MMDiT block: separate img/txt params, joint
    attention over the concatenated sequence."""

    def __init__(self, hidden: int, num_heads: int, mlp_ratio: float = 4.0):
        super().__init__()
        head_dim = hidden // num_heads
        mlp_dim = int(hidden * mlp_ratio)
        self.num_heads = num_heads
        self.img_mod = Modulation(hidden, double=True)
        self.img_attn_qkv = nn.Linear(hidden, hidden * 3)
        self.img_attn_norm = QKNorm(head_dim)
        self.img_attn_proj = nn.Linear(hidden, hidden)
        self.img_mlp = FusedMLP(hidden, mlp_dim, hidden)
        self.txt_mod = Modulation(hidden, double=True)
        self.txt_attn_qkv = nn.Linear(hidden, hidden * 3)
        self.txt_attn_norm = QKNorm(head_dim)
        self.txt_attn_proj = nn.Linear(hidden, hidden)
        self.txt_mlp = FusedMLP(hidden, mlp_dim, hidden)
        self.attn = JointAttention(num_heads, head_dim)

    def _qkv(self, x, qkv_layer, norm, pe_slice):
        """Fused-qkv projection -> strided [B,S,H,D] views (no transposes),
        with in-place fused qk RMSNorm + RoPE on the projection buffer."""
        B, S, _ = x.shape
        qkv = qkv_layer(x).unflatten(-1, (3, self.num_heads, -1))
        q, k, v = qkv.unbind(2)
        ops.qk_norm_rope_(
            q, k, norm.query_norm.scale, norm.key_norm.scale, pe_slice
        )
        return q, k, v

    def forward(self, img, txt, vec, pe, mods=None):
        if mods is not None:
            (img_m1, img_m2), (txt_m1, txt_m2) = mods
        else:
            img_m1, img_m2 = self.img_mod(vec)
            txt_m1, txt_m2 = self.txt_mod(vec)

        T = txt.shape[1]
        H = self.num_heads
        txt_qkv = ln_mod_into(
            self.txt_attn_qkv, txt, txt_m1.scale, txt_m1.shift
        ).unflatten(-1, (3, H, -1))
        img_qkv = ln_mod_into(
            self.img_attn_qkv, img, img_m1.scale, img_m1.shift
        ).unflatten(-1, (3, H, -1))
        # fused: per-stream qk-norm + RoPE + contiguous joint q/k/v
        # (txt first, then img — FLUX convention)
        q, k, v = ops.pack_joint_qkv(
            txt_qkv, img_qkv,
            self.txt_attn_norm.query_norm.scale,
            self.txt_attn_norm.key_norm.scale,
            self.img_attn_norm.query_norm.scale,
            self.img_attn_norm.key_norm.scale,
            pe,
        )
        txt_attn, img_attn = ops.attention_bshd_split(
            q, k, v, T, self.attn.scale
        )
        txt_attn = txt_attn.flatten(2)
        img_attn = img_attn.flatten(2)

        img = ops.gate_residual(img, img_m1.gate, self.img_attn_proj(img_attn))
        img = ops.gate_residual(
            img, img_m2.gate,
            self.img_mlp.forward_ln(img, img_m2.scale, img_m2.shift),
        )
        txt = ops.gate_residual(txt, txt_m1.gate, self.txt_attn_proj(txt_attn))
        txt = ops.gate_residual(
            txt, txt_m2.gate,
            self.txt_mlp.forward_ln(txt, txt_m2.scale, txt_m2.shift),
        )
        return img, txt


class SingleStreamBlock(nn.Module):
    """FLUX-class single-stream block: fused qkv+mlp in, attn ∥ mlp, fused out."""

    def __init__(self, hidden: int, num_heads: int, mlp_ratio: float = 4.0):
        super().__init__()
        head_dim = hidden // num_heads
        self.num_heads = num_heads
        self.mlp_dim = int(hidden * mlp_ratio)
        # split projections: qkv separately from the MLP up-projection so
        # the GELU fuses into the MLP GEMM epilogue; the output projection is
        # split the same way (no strided last-dim concats anywhere)
        self.linear1_qkv = nn.Linear(hidden, hidden * 3)
        self.linear1_mlp = GELULinear(hidden, self.mlp_dim)
        self.linear2_attn = nn.Linear(hidden, hidden)
        self.linear2_mlp = nn.Linear(self.mlp_dim, hidden, bias=False)
        self.norm = QKNorm(head_dim)
        self.modulation = Modulation(hidden, double=False)
        self.attn = JointAttention(num_heads, head_dim)

    def forward(self, x, vec, pe, mods=None):
        B, S, hidden = x.shape
        if mods is not None:
            (mod,) = mods
        else:
            mod, _ = self.modulation(vec)
        qkv_lin = self.linear1_qkv
        mlp_lin = getattr(self.linear1_mlp, "lin", None)
        mlp_fp8 = None  # (x8, scale) for the fused gelu+quant hand-off
        if (hasattr(qkv_lin, "ln_quant") and hasattr(mlp_lin, "mm_fp8")
                and x.is_cuda and x.dtype == torch.bfloat16
                and ops.hip_available("layer_norm_mod_fp8")):
            # fp8 mode: ONE fused LN+quant pass shared by both consumers
            # (qkv and mlp-up read the same normalized activation) — was
            # one LN + two standalone quant passes.
            x8, s_used = qkv_lin.ln_quant(x, mod.scale, mod.shift)
            qkv = qkv_lin.mm_fp8(x8, s_used)
            y_up = mlp_lin.mm_fp8(x8, s_used)
            if (hasattr(self.linear2_mlp, "gelu_quant")
                    and ops.hip_available("gelu_fp8")):
                # GELU + e4m3 cast in one kernel feeding linear2_mlp
                mlp_fp8 = self.linear2_mlp.gelu_quant(y_up)
                mlp_act = None
            else:
                mlp_act = ops.gelu_tanh(y_up)
        else:
            x_in = ops.layer_norm_mod(x, mod.scale, mod.shift)
            qkv = self.linear1_qkv(x_in)
            mlp_act = self.linear1_mlp(x_in)  # GELU in the GEMM epilogue
        qkv = qkv.unflatten(-1, (3, self.num_heads, -1))
        q, k, v = qkv.unbind(2)  # [B,S,H,D] views
        ops.qk_norm_rope_(
            q, k, self.norm.query_norm.scale, self.norm.key_norm.scale, pe
        )
        attn = ops.attention_bshd(q, k, v, self.attn.scale).flatten(2)
        if mlp_fp8 is not None:
            out = self.linear2_attn(attn) + self.linear2_mlp.mm_fp8(*mlp_fp8)
        elif x.is_cuda and isinstance(self.linear2_mlp, nn.Linear):
            # second GEMM accumulates into the first's output (beta=1
            # epilogue) — no separate elementwise add
            acc = self.linear2_attn(attn).reshape(-1, hidden)
            out = torch.addmm(
                acc, mlp_act.reshape(-1, self.mlp_dim),
                self.linear2_mlp.weight.t(),
            ).reshape(B, S, hidden)
        else:
            out = self.linear2_attn(attn) + self.linear2_mlp(mlp_act)
        return ops.gate_residual(x, mod.gate, out)


class LastLayer(nn.Module):
    """Final AdaLN + projection to patch output."""

    def __init__(self, hidden: int, out_dim: int):
        super().__init__()
        self.ada_lin = nn.Linear(hidden, 2 * hidden, bias=True)
        self.linear = nn.Linear(hidden, out_dim, bias=True)

    def forward(self, x, vec):
        shift, scale = self.ada_lin(torch.nn.functional.silu(vec)).chunk(2, dim=-1)
        return self.linear(ops.layer_norm_mod(x, scale, shift))


def _mod_outs(chunked):
    """list of 3-chunk groups -> ModOut tuples."""
    return ModOut(*chunked[:3]), (ModOut(*chunked[3:]) if len(chunked) == 6 else None)


class ModulationBank:
    """One fused GEMM for every per-block AdaLN projection.

    All Modulation.lin layers (and nothing else) consume silu(vec), so their
    weights concatenate into a single [sum_out, hidden] matrix: one
    hipBLASLt call per step replaces ~115 tiny GEMMs + silu launches. The
    concatenated weights are a per-replica cache (CACHE_ATTRS clears it on
    replication; it rebuilds lazily from the replica's own - possibly
    LoRA-patched - weights on its own device).
    """

    def __init__(self, mod_modules):
        self.mods = list(mod_modules)  # Modulation instances, block order
        if not all(isinstance(m.lin, nn.Linear) for m in self.mods):
            # quantized/wrapped projections: banking disabled, blocks
            # compute their own modulation (mods=None path)
            raise TypeError("ModulationBank requires plain nn.Linear mods")
        self.weight = None
        self.bias = None
        self.splits = [m.lin.out_features for m in self.mods]

    def _build(self):
        self.weight = torch.cat([m.lin.weight for m in self.mods], dim=0)
        self.bias = torch.cat([m.lin.bias for m in self.mods], dim=0)

    def __call__(self, vec):
        if self.weight is None or self.weight.device != vec.device:
            self._build()
        out = torch.nn.functional.linear(
            torch.nn.functional.silu(vec), self.weight, self.bias
        )
        results = []
        off = 0
        for m, width in zip(self.mods, self.splits):
            seg = out[:, off:off + width]
            off += width
            dim = m.lin.in_features
            parts = [seg[:, i * dim:(i + 1) * dim] for i in range(m.multiplier)]
            results.append(_mod_outs(parts))
        return results
