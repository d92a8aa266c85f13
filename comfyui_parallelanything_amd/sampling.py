"""Headless samplers: the denoising loops ComfyUI's KSampler provides
upstream of the reference node.

The engine's contract is one model forward per denoising iteration
(reference monkeypatched forward, any_device_parallel.py:1287); these
drivers call the (possibly parallel-installed) model once per step:

- flow-matching Euler / Heun (FLUX, Z-Image, WAN class: model predicts
  velocity; x moves along sigma from 1 -> 0),
- DPM++ 2M for epsilon-prediction UNets (SD1.5/SDXL class), Karras sigmas.
"""
from __future__ import annotations

import math
from typing import Callable, Optional

import torch


def flow_sigmas(steps: int, shift: float = 1.0, device="cpu") -> torch.Tensor:
    """Linear flow schedule 1 -> 0 with optional resolution shift
    (sigma' = shift*s / (1 + (shift-1)*s), FLUX convention)."""
    s = torch.linspace(1.0, 0.0, steps + 1, device=device)
    if shift != 1.0:
        s = shift * s / (1.0 + (shift - 1.0) * s)
    return s


def karras_sigmas(steps: int, sigma_min: float = 0.0292,
                  sigma_max: float = 14.61, rho: float = 7.0,
                  device="cpu") -> torch.Tensor:
    ramp = torch.linspace(0, 1, steps, device=device)
    min_r, max_r = sigma_min ** (1 / rho), sigma_max ** (1 / rho)
    sig = (max_r + ramp * (min_r - max_r)) ** rho
    return torch.cat([sig, torch.zeros(1, device=device)])


@torch.no_grad()
def sample_flow_euler(model: Callable, x: torch.Tensor, sigmas: torch.Tensor,
                      context=None, callback: Optional[Callable] = None,
                      **kwargs) -> torch.Tensor:
    """Euler over the flow ODE dx/ds = v(x, s)."""
    B = x.shape[0]
    for i in range(len(sigmas) - 1):
        t = torch.full((B,), float(sigmas[i]), device=x.device)
        v = model(x, t, context=context, **kwargs)
        x = x + (sigmas[i + 1] - sigmas[i]) * v.to(x.dtype)
        if callback:
            callback(i, x)
    return x


@torch.no_grad()
def sample_flow_heun(model: Callable, x: torch.Tensor, sigmas: torch.Tensor,
                     context=None, **kwargs) -> torch.Tensor:
    """Heun (2nd order) over the flow ODE; two model calls per step."""
    B = x.shape[0]
    for i in range(len(sigmas) - 1):
        s0, s1 = float(sigmas[i]), float(sigmas[i + 1])
        h = s1 - s0
        t0 = torch.full((B,), s0, device=x.device)
        v0 = model(x, t0, context=context, **kwargs).to(x.dtype)
        x_pred = x + h * v0
        if s1 == 0.0:
            x = x_pred
        else:
            t1 = torch.full((B,), s1, device=x.device)
            v1 = model(x_pred, t1, context=context, **kwargs).to(x.dtype)
            x = x + h * 0.5 * (v0 + v1)
    return x


@torch.no_grad()
def sample_dpmpp_2m(model: Callable, x: torch.Tensor, sigmas: torch.Tensor,
                    context=None, **kwargs) -> torch.Tensor:
    """DPM-Solver++(2M) for epsilon-prediction models (Karras sigmas).

    denoised D = x - sigma * eps; standard 2M multistep update in
    log-sigma time.
    """
    B = x.shape[0]
    old_d = None
    for i in range(len(sigmas) - 1):
        sig, sig_next = float(sigmas[i]), float(sigmas[i + 1])
        t = torch.full((B,), sig, device=x.device)
        eps = model(x, t, context=context, **kwargs).to(x.dtype)
        denoised = x - sig * eps
        if sig_next == 0.0:
            x = denoised
        else:
            lt, lt_next = math.log(sig), math.log(sig_next)
            h = lt_next - lt
            if old_d is None:
                x = (sig_next / sig) * x - torch.expm1(
                    torch.tensor(h)).item() * denoised
            else:
                h_last = lt - math.log(float(sigmas[i - 1]))
                r = h_last / h
                d = (1 + 1 / (2 * r)) * denoised - (1 / (2 * r)) * old_d
                x = (sig_next / sig) * x - torch.expm1(
                    torch.tensor(h)).item() * d
            old_d = denoised
    return x


SAMPLERS = {
    "euler": sample_flow_euler,
    "heun": sample_flow_heun,
    "dpmpp_2m": sample_dpmpp_2m,
}
