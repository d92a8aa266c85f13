import pytest
import torch

from comfyui_parallelanything_amd.parallel.chain import (
    DeviceChain,
    available_devices,
    chain_append,
    chain_from_slots,
    make_entry,
    normalize_weights,
)


def test_entry_schema():
    e = make_entry("cpu", 30.0)
    assert e == {"device": "cpu", "percentage": 30.0, "weight": 0.3}


def test_chain_append_preserves_prev():
    c1 = chain_append(None, "cpu", 40)
    c2 = chain_append(c1, "cpu", 60)
    assert len(c1) == 1 and len(c2) == 2
    assert c2[1]["percentage"] == 60.0


def test_slots_drop_zero_percent():
    chain = chain_from_slots([("cpu", 50), ("cpu", 0), ("cpu", 25), ("cpu", 0)])
    assert [e["percentage"] for e in chain] == [50.0, 25.0]


def test_normalize_not_summing_to_100():
    # reference renormalizes (any_device_parallel.py:1019-1027)
    chain = [make_entry("cpu", 30), make_entry("cpu", 30)]
    assert normalize_weights(chain) == [0.5, 0.5]
    chain = [make_entry("cpu", 150), make_entry("cpu", 50)]
    assert normalize_weights(chain) == [0.75, 0.25]


def test_normalize_zero_total_even_split():
    chain = [{"device": "cpu", "percentage": 0}, {"device": "cpu", "percentage": 0}]
    assert normalize_weights(chain) == [0.5, 0.5]


def test_device_chain_lead_and_drop():
    dc = DeviceChain.from_list(
        [make_entry("cpu", 60), make_entry("cpu", 30), make_entry("cpu", 10)]
    )
    assert dc.lead == "cpu" and len(dc) == 3
    d2 = dc.drop(1)
    assert len(d2) == 2
    assert d2.weights[0] == pytest.approx(60 / 70)
    assert d2.weights[1] == pytest.approx(10 / 70)


def test_device_chain_empty_raises():
    with pytest.raises(ValueError):
        DeviceChain.from_list([])


def test_device_chain_bad_device_raises():
    with pytest.raises(Exception):
        DeviceChain.from_list([make_entry("not_a_device", 100)])


def test_available_devices_cpu_first():
    devs = available_devices()
    assert devs[0] == "cpu"
    if torch.cuda.is_available():
        assert "cuda:0" in devs
