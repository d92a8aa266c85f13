"""Replication-ladder tests: hostile foreign modules, cache scrub rules.

The reference survives arbitrary ComfyUI diffusion_model classes via a
two-strategy clone ladder (any_device_parallel.py:586-672 + :390-584) and
scrubs ~24 device-bound cache attr names (:167-174). These tests pin our
equivalents: deepcopy-with-memo first, recursive structural clone on
failure, and a cache scrub that nulls plain tensor attrs with the
reference's names WITHOUT destroying registered parameters/buffers.
"""
import threading

import pytest
import torch
from torch import nn

from comfyui_parallelanything_amd.parallel.replicate import (
    FOREIGN_CACHE_ATTRS,
    clear_replica_caches,
    replicate_module,
)


class HostileInner(nn.Module):
    """Submodule with a non-deepcopyable attr and a device-bound cache."""

    def __init__(self):
        super().__init__()
        self.lin = nn.Linear(8, 8)
        self._lock = threading.Lock()       # deepcopy raises TypeError
        self.freqs_cis = None               # populated lazily per device

    def forward(self, x):
        if self.freqs_cis is None or self.freqs_cis.device != x.device:
            self.freqs_cis = torch.arange(
                8, dtype=x.dtype, device=x.device
            ).sin()
        return self.lin(x) + self.freqs_cis


class HostileModel(nn.Module):
    def __init__(self):
        super().__init__()
        self.inner = HostileInner()
        self.head = nn.Linear(8, 4)
        self.scale_factor = 0.5             # plain attr must survive clone
        self.img_ids = torch.zeros(3, 3)    # foreign cache attr (plain)

    def forward(self, x):
        return self.head(self.inner(x)) * self.scale_factor


def test_hostile_module_structural_fallback():
    """VERDICT round-1 'Done =': non-deepcopyable attr + a freqs_cis tensor
    cached on the source device -> correct, cache-clean replica."""
    torch.manual_seed(0)
    m = HostileModel()
    x = torch.randn(2, 8)
    ref = m(x)                              # populates inner.freqs_cis
    assert m.inner.freqs_cis is not None

    rep = replicate_module(m, "cpu", force_copy=True)
    assert rep is not m
    # caches scrubbed on the replica (repopulate on its own device)...
    assert rep.inner.freqs_cis is None
    assert rep.img_ids is None
    # ...and the SOURCE is untouched
    assert m.inner.freqs_cis is not None
    # parameters are copies, not aliases
    assert rep.inner.lin.weight.data_ptr() != m.inner.lin.weight.data_ptr()
    assert not rep.inner.lin.weight.requires_grad
    # plain attrs came across; uncopyable attrs exist (shared, best-effort)
    assert rep.scale_factor == 0.5
    assert hasattr(rep.inner, "_lock")
    # forward equality
    out = rep(x)
    torch.testing.assert_close(out, ref)


def test_hostile_replica_independent_after_source_mutation():
    m = HostileModel()
    x = torch.randn(2, 8)
    rep = replicate_module(m, "cpu", force_copy=True)
    with torch.no_grad():
        m.inner.lin.weight.zero_()
    assert rep.inner.lin.weight.abs().sum() > 0  # replica unaffected


class SharedTail(nn.Module):
    """Two names referencing the SAME submodule (weight tying)."""

    def __init__(self):
        super().__init__()
        self.a = nn.Linear(4, 4)
        self.b = self.a
        self._lock = threading.Lock()  # forces the structural path

    def forward(self, x):
        return self.b(self.a(x))


def test_structural_clone_preserves_module_aliasing():
    m = SharedTail()
    rep = replicate_module(m, "cpu", force_copy=True)
    assert rep.a is rep.b, "tied submodules must stay tied in the replica"
    x = torch.randn(2, 4)
    torch.testing.assert_close(rep(x), m(x))


class RegisteredPosEmbed(nn.Module):
    """pos_embed as a registered Parameter is a WEIGHT, not a cache."""

    def __init__(self):
        super().__init__()
        self.pos_embed = nn.Parameter(torch.randn(4, 4))
        self.register_buffer("freqs", torch.randn(4))
        self.rope_cache = torch.randn(4)    # plain attr: a true cache

    def forward(self, x):
        return x + self.pos_embed + self.freqs + 0 * self.rope_cache.sum()


def test_scrub_never_touches_registered_params_or_buffers():
    m = RegisteredPosEmbed()
    n = clear_replica_caches(m)
    assert n == 1
    assert m.pos_embed is not None          # registered Parameter kept
    assert m.freqs is not None              # registered buffer kept
    assert m.rope_cache is None             # plain tensor attr scrubbed


def test_scrub_on_deepcopy_path_too():
    """Foreign caches are scrubbed even when deepcopy succeeds."""

    class Clean(nn.Module):
        def __init__(self):
            super().__init__()
            self.lin = nn.Linear(4, 4)
            self.txt_ids = torch.zeros(2, 2)

    m = Clean()
    rep = replicate_module(m, "cpu", force_copy=True)
    assert rep.txt_ids is None
    assert m.txt_ids is not None


def test_foreign_attr_list_covers_reference_names():
    """The scrub list must cover the reference's clear_flux_caches names
    (any_device_parallel.py:167-174)."""
    for name in (
        "freqs_cis", "img_ids", "txt_ids", "pos_embed", "kv_cache",
        "temporal_ids", "attn_bias", "rope_cache", "frame_ids",
    ):
        assert name in FOREIGN_CACHE_ATTRS


def test_dict_of_tensors_cache_cleared():
    class DictCache(nn.Module):
        def __init__(self):
            super().__init__()
            self.lin = nn.Linear(2, 2)
            self.cache = {"k": torch.zeros(2)}

    m = DictCache()
    clear_replica_caches(m)
    assert m.cache == {}


def test_structural_clone_fuzz_attr_types():
    """Hypothesis-style fuzz (deterministic grid) of the structural path:
    modules carrying many plain-attr shapes must clone correctly."""
    import threading

    payloads = [
        3, 3.5, "s", b"b", None, True,
        [1, 2, 3], (4, 5), {"a": 1}, {"t": torch.ones(3)},
        torch.arange(6).reshape(2, 3), [torch.zeros(2), torch.ones(2)],
        {"nested": {"deep": [torch.full((2,), 7.0)]}},
        range(5), frozenset({1, 2}),
    ]
    for i, payload in enumerate(payloads):
        class M(nn.Module):
            def __init__(self):
                super().__init__()
                self.lin = nn.Linear(4, 4)
                self._lock = threading.Lock()   # force structural path
                self.payload = payload

            def forward(self, x):
                return self.lin(x)

        torch.manual_seed(i)
        m = M()
        x = torch.randn(2, 4)
        ref = m(x)
        rep = replicate_module(m, "cpu", force_copy=True)
        torch.testing.assert_close(rep(x), ref)
        # tensors in payloads become copies; scalars/strings survive
        if isinstance(payload, torch.Tensor):
            assert torch.equal(rep.payload, payload)
            assert rep.payload.data_ptr() != payload.data_ptr()
        elif isinstance(payload, (int, float, str, bytes, bool)) or payload is None:
            assert rep.payload == payload or rep.payload is payload
        # replica params independent of the source
        with torch.no_grad():
            m.lin.weight.add_(1.0)
        torch.testing.assert_close(rep(x), ref)
