from collections import OrderedDict

import torch

from torchsnapshot_amd.flatten import (
    escape_key,
    flatten,
    inflate,
    unescape_key,
)
from torchsnapshot_amd.manifest import DictEntry, ListEntry, OrderedDictEntry


def test_escape_round_trip():
    for key in ["plain", "has/slash", "has%percent", "%2F", "a/b%c/d", ""]:
        assert unescape_key(escape_key(key)) == key


def test_flatten_inflate_round_trip():
    obj = {
        "a": 1,
        "b": {"c": [1, 2, {"d": torch.tensor([1.0, 2.0])}], "e": "str"},
        "od": OrderedDict([("z", 1), ("a", 2)]),
        "weird/key": {"x%y": 3.5},
        5: "int key",
        "empty": {},
        "empty_list": [],
    }
    manifest, flattened = flatten(obj, prefix="root")
    rebuilt = inflate(manifest, flattened, prefix="root")
    assert rebuilt.keys() == obj.keys()
    assert rebuilt["a"] == 1
    assert torch.equal(rebuilt["b"]["c"][2]["d"], obj["b"]["c"][2]["d"])
    assert isinstance(rebuilt["od"], OrderedDict)
    assert list(rebuilt["od"].keys()) == ["z", "a"]
    assert rebuilt["weird/key"]["x%y"] == 3.5
    assert rebuilt[5] == "int key"
    assert rebuilt["empty"] == {}
    assert rebuilt["empty_list"] == []


def test_flatten_entry_types():
    obj = {"d": {}, "l": [], "od": OrderedDict()}
    manifest, flattened = flatten(obj, prefix="p")
    assert isinstance(manifest["p"], DictEntry)
    assert isinstance(manifest["p/d"], DictEntry)
    assert isinstance(manifest["p/l"], ListEntry)
    assert isinstance(manifest["p/od"], OrderedDictEntry)
    assert flattened == {}


def test_ambiguous_keys_become_leaf():
    # int 1 and str "1" escape identically: container must not be flattened
    obj = {"amb": {1: "a", "1": "b"}}
    manifest, flattened = flatten(obj, prefix="p")
    assert "p/amb" in flattened
    assert flattened["p/amb"] == {1: "a", "1": "b"}


def test_non_str_int_keys_become_leaf():
    obj = {"t": {(1, 2): "tuple-key"}}
    manifest, flattened = flatten(obj, prefix="p")
    assert flattened["p/t"] == {(1, 2): "tuple-key"}


def test_leaf_paths_escaped():
    obj = {"a/b": 1}
    manifest, flattened = flatten(obj, prefix="p")
    assert "p/a%2Fb" in flattened


def test_nested_list_ordering():
    obj = {"l": list(range(15))}
    manifest, flattened = flatten(obj, prefix="p")
    rebuilt = inflate(manifest, flattened, prefix="p")
    assert rebuilt["l"] == list(range(15))
