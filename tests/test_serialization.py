"""CPU-portability payload helper (`unionml_amd/utils/serialization.py`):
the backend moves tensor payloads to CPU at process boundaries. These
tests pin the container semantics that matter for artifacts crossing
the execution directory (no GPU required: CPU objects must pass through
untouched, containers must keep their types)."""

import collections

import torch

from unionml_amd.utils.serialization import tensors_to_cpu


def test_cpu_tensor_passes_through_same_object():
    t = torch.randn(4)
    assert tensors_to_cpu(t) is t


def test_cpu_module_passes_through_same_object():
    m = torch.nn.Linear(3, 2)
    assert tensors_to_cpu(m) is m  # no deepcopy cost for CPU modules


def test_nested_containers_recurse_and_keep_types():
    t = torch.randn(2)
    obj = {"a": [t, (t, {"b": t})], "c": 7, "d": "s"}
    out = tensors_to_cpu(obj)
    assert isinstance(out["a"], list) and isinstance(out["a"][1], tuple)
    assert out["c"] == 7 and out["d"] == "s"
    assert out["a"][0] is t  # CPU tensors untouched even when nested


def test_namedtuple_type_preserved():
    Split = collections.namedtuple("Split", ["train", "test"])
    s = Split(train=torch.randn(3), test=torch.randn(2))
    out = tensors_to_cpu(s)
    assert type(out) is Split
    assert out.train is s.train and out.test is s.test


def test_non_tensor_objects_untouched():
    class Opaque:
        pass

    o = Opaque()
    assert tensors_to_cpu(o) is o
    assert tensors_to_cpu(None) is None
