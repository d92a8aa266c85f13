import pytest
import torch

from comfyui_parallelanything_amd.parallel.split import (
    active_split,
    compute_split_sizes,
    concatenate_results,
    get_batch_size,
    move_to_device,
    split_batch,
    split_kwargs,
)


# --- split sizes (reference semantics: any_device_parallel.py:1321-1322) ---

def test_split_even():
    assert compute_split_sizes(8, [0.5, 0.5]) == [4, 4]


def test_split_weighted_remainder_to_last():
    # 21 * 0.6 = 12.6 -> 12; last gets 21-12 = 9 (the README headline split)
    assert compute_split_sizes(21, [0.6, 0.4]) == [12, 9]


def test_split_min_one_floor():
    # int(10*0.05) = 0 -> floored to 1
    assert compute_split_sizes(10, [0.05, 0.95]) == [1, 9]


def test_split_sum_invariant():
    for b in range(1, 40):
        for w in ([0.7, 0.3], [0.25, 0.25, 0.5], [0.9, 0.05, 0.05]):
            sizes = compute_split_sizes(b, w)
            assert sum(sizes) == b
            assert all(s >= 0 for s in sizes)


def test_split_negative_tail_repaired():
    # 4 devices, batch 3: min-1 floors over-commit; tail must not go negative
    sizes = compute_split_sizes(3, [0.25] * 4)
    assert sum(sizes) == 3 and all(s >= 0 for s in sizes)


def test_active_split_drops_zeros():
    devs, ws, ss = active_split(["a", "b", "c"], [0.4, 0.3, 0.3], [2, 0, 1])
    assert devs == ["a", "c"] and ss == [2, 1]


def test_active_split_all_zero_raises():
    with pytest.raises(ValueError):
        active_split(["a"], [1.0], [0])


# --- batch detection (reference :1210-1220) ---

def test_batch_size_tensor():
    assert get_batch_size(torch.zeros(5, 3)) == 5


def test_batch_size_list_of_tensors():
    assert get_batch_size([torch.zeros(7, 2), torch.zeros(7, 4)]) == 7


def test_batch_size_list_no_tensors():
    assert get_batch_size(["a", "b", "c"]) == 3


def test_batch_size_scalar():
    assert get_batch_size(42) == 1


# --- split_batch (reference :1222-1237) ---

def test_split_batch_tensor():
    chunks = split_batch(torch.arange(10).view(10, 1), [6, 4])
    assert [c.shape[0] for c in chunks] == [6, 4]
    assert chunks[1][0].item() == 6


def test_split_batch_list_mixed():
    x = [torch.zeros(4, 2), "meta"]
    chunks = split_batch(x, [1, 3])
    assert chunks[0][0].shape[0] == 1 and chunks[1][0].shape[0] == 3
    assert chunks[0][1] == "meta" and chunks[1][1] == "meta"


def test_split_batch_non_tensor_broadcast():
    assert split_batch("cond", [2, 2]) == ["cond", "cond"]


# --- split_kwargs (reference :1252-1267) ---

def test_split_kwargs_batch_tensor_splits():
    kw = {"mask": torch.zeros(6, 3), "flag": True}
    out = split_kwargs(kw, [2, 4], 6)
    assert out[0]["mask"].shape[0] == 2 and out[1]["mask"].shape[0] == 4
    assert out[0]["flag"] is True and out[1]["flag"] is True


def test_split_kwargs_nonbatch_tensor_broadcast():
    # tensor whose dim0 != batch is broadcast whole
    kw = {"table": torch.zeros(3, 3)}
    out = split_kwargs(kw, [2, 4], 6)
    assert out[0]["table"].shape == (3, 3) and out[1]["table"].shape == (3, 3)


def test_split_kwargs_list_of_batch_tensors():
    kw = {"feats": [torch.zeros(6, 2), torch.zeros(6, 4)]}
    out = split_kwargs(kw, [1, 5], 6)
    assert out[0]["feats"][0].shape[0] == 1
    assert out[1]["feats"][1].shape[0] == 5
    assert isinstance(out[0]["feats"], list)


def test_split_kwargs_mixed_list_broadcast():
    kw = {"feats": [torch.zeros(6, 2), "not_a_tensor"]}
    out = split_kwargs(kw, [3, 3], 6)
    assert out[0]["feats"] is kw["feats"]  # broadcast whole


# --- concat (reference :1269-1285) ---

def test_concat_tensors():
    out = concatenate_results([torch.ones(2, 3), torch.zeros(3, 3)])
    assert out.shape == (5, 3)


def test_concat_nested_tuple():
    r1 = (torch.ones(2, 3), "aux")
    r2 = (torch.zeros(1, 3), "aux")
    out = concatenate_results([r1, r2])
    assert isinstance(out, tuple)
    assert out[0].shape == (3, 3) and out[1] == "aux"


def test_concat_empty():
    assert concatenate_results([]) == []


# --- move_to_device ---

def test_move_to_device_recursive():
    x = {"a": torch.zeros(2), "b": [torch.ones(1), 5]}
    out = move_to_device(x, "cpu")
    assert out["b"][1] == 5 and out["a"].device.type == "cpu"
