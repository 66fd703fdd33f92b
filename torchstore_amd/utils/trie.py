"""Separator-aware prefix trie used by the controller's key index.

Replaces the reference's pygtrie dependency (torchstore
``storage_utils/trie.py``, a ``StringTrie`` with separator ``"."``) with a
small self-contained implementation.  Keys are tokenized on BOTH ``"/"``
(the hierarchical ``"{key}/{flat_key}"`` layout the state_dict layer
writes) and ``"."`` (the dotted module paths inside a flat key), with each
separator kept attached to the component it introduces — so ``"a.b"`` and
``"a/b"`` are distinct keys, and ``keys(prefix=...)`` matches whole
components across both conventions: prefix ``"sd/model"`` matches
``"sd/model.layers.0.weight"`` but not ``"sd/model2"`` (the reference's
dotted-prefix behavior users of ``ts.keys`` expect).
"""

from __future__ import annotations

from typing import Any, Dict, Iterator, List, MutableMapping, Optional, Tuple

_MISSING = object()
_SEPARATORS = ("/", ".")


class _Node:
    __slots__ = ("children", "value", "has_value")

    def __init__(self):
        self.children: Dict[str, "_Node"] = {}
        self.value: Any = None
        self.has_value = False


class Trie(MutableMapping):
    def __init__(self, separators: Tuple[str, ...] = _SEPARATORS):
        self._seps = frozenset(separators)
        self._root = _Node()
        self._len = 0

    # -- helpers ---------------------------------------------------------
    def _parts(self, key: str) -> List[str]:
        """Tokenize with each separator ATTACHED to the part it opens:
        ``"a/b.c"`` → ``["a", "/b", ".c"]`` — concatenation reconstructs
        the exact key, and ``"a.b"`` / ``"a/b"`` stay distinct."""
        if not isinstance(key, str):
            raise TypeError(f"trie keys are strings, got {type(key)}")
        parts: List[str] = []
        cur: List[str] = []
        for ch in key:
            if ch in self._seps and cur:
                parts.append("".join(cur))
                cur = [ch]
            else:
                cur.append(ch)
        parts.append("".join(cur))
        return parts

    def _find(self, key: str) -> Optional[_Node]:
        node = self._root
        for p in self._parts(key):
            node = node.children.get(p)
            if node is None:
                return None
        return node

    # -- MutableMapping --------------------------------------------------
    def __setitem__(self, key: str, value: Any) -> None:
        node = self._root
        for p in self._parts(key):
            node = node.children.setdefault(p, _Node())
        if not node.has_value:
            self._len += 1
        node.value = value
        node.has_value = True

    def __getitem__(self, key: str) -> Any:
        node = self._find(key)
        if node is None or not node.has_value:
            raise KeyError(key)
        return node.value

    def __delitem__(self, key: str) -> None:
        parts = self._parts(key)
        path: List[Tuple[_Node, str]] = []
        node = self._root
        for p in parts:
            nxt = node.children.get(p)
            if nxt is None:
                raise KeyError(key)
            path.append((node, p))
            node = nxt
        if not node.has_value:
            raise KeyError(key)
        node.has_value = False
        node.value = None
        self._len -= 1
        # prune empty branches
        for parent, part in reversed(path):
            child = parent.children[part]
            if child.has_value or child.children:
                break
            del parent.children[part]

    def __contains__(self, key: object) -> bool:
        if not isinstance(key, str):
            return False
        node = self._find(key)
        return node is not None and node.has_value

    def __len__(self) -> int:
        return self._len

    def __iter__(self) -> Iterator[str]:
        yield from self._iter(self._root, [])

    def _iter(self, node: _Node, parts: List[str]) -> Iterator[str]:
        if node.has_value:
            yield "".join(parts)
        for part in node.children:
            parts.append(part)
            yield from self._iter(node.children[part], parts)
            parts.pop()

    # -- prefix queries ---------------------------------------------------
    def keys_with_prefix(self, prefix: Optional[str] = None) -> List[str]:
        """All keys whose leading path components equal ``prefix``'s."""
        if prefix is None or prefix == "":
            return list(self)
        node = self._find(prefix)
        if node is None:
            return []
        return list(self._iter(node, self._parts(prefix)))

    def pop(self, key: str, default: Any = _MISSING) -> Any:
        try:
            value = self[key]
        except KeyError:
            if default is _MISSING:
                raise
            return default
        del self[key]
        return value
