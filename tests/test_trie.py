"""Prefix-trie key index."""

import pytest

from torchstore_amd.utils.trie import Trie


def test_basic_mapping():
    t = Trie()
    t["a/b/c"] = 1
    t["a/b"] = 2
    t["x"] = 3
    assert t["a/b/c"] == 1
    assert t["a/b"] == 2
    assert len(t) == 3
    assert "a/b" in t
    assert "a" not in t  # intermediate node, no value
    assert sorted(t) == ["a/b", "a/b/c", "x"]


def test_delete_and_prune():
    t = Trie()
    t["m/l1/w"] = 1
    t["m/l2/w"] = 2
    del t["m/l1/w"]
    assert "m/l1/w" not in t
    assert t["m/l2/w"] == 2
    with pytest.raises(KeyError):
        del t["m/l1/w"]
    with pytest.raises(KeyError):
        _ = t["nope"]


def test_prefix_component_matching():
    t = Trie()
    t["model/layer.0"] = 1
    t["model/layer.1"] = 2
    t["model2/x"] = 3
    keys = sorted(t.keys_with_prefix("model"))
    assert keys == ["model/layer.0", "model/layer.1"]
    assert t.keys_with_prefix("model/layer.0") == ["model/layer.0"]
    assert t.keys_with_prefix("mod") == []
    assert sorted(t.keys_with_prefix(None)) == sorted(t)


def test_overwrite_and_pop():
    t = Trie()
    t["k"] = 1
    t["k"] = 2
    assert t["k"] == 2 and len(t) == 1
    assert t.pop("k") == 2
    assert t.pop("k", "d") == "d"
    with pytest.raises(KeyError):
        t.pop("k")


def test_dotted_prefix_matching():
    """Reference parity: pygtrie StringTrie(separator=".") filters dotted
    module paths by component (torchstore storage_utils/trie.py)."""
    t = Trie()
    t["sd/model.layers.0.weight"] = 1
    t["sd/model.layers.1.weight"] = 2
    t["sd/model2.weight"] = 3
    t["sd/<MAPPING>"] = 4
    # dotted prefix descends "." components
    assert sorted(t.keys_with_prefix("sd/model.layers")) == [
        "sd/model.layers.0.weight",
        "sd/model.layers.1.weight",
    ]
    # "sd/model" must NOT match "sd/model2" (component boundary)
    assert sorted(t.keys_with_prefix("sd/model")) == [
        "sd/model.layers.0.weight",
        "sd/model.layers.1.weight",
    ]
    assert len(t.keys_with_prefix("sd")) == 4


def test_mixed_separators_stay_distinct():
    t = Trie()
    t["a.b"] = 1
    t["a/b"] = 2
    assert t["a.b"] == 1
    assert t["a/b"] == 2
    assert len(t) == 2
    del t["a.b"]
    assert "a.b" not in t and t["a/b"] == 2
