"""Attribute-access config dict (replaces the reference's EasyDict dependency).

The reference loads YAML into ``easydict.EasyDict`` (reference main.py:113-115)
and accesses config values as attributes throughout. ``easydict`` is not part
of this image, so we own a minimal recursive implementation with the same
access semantics (attribute get/set, nested dict conversion, normal dict API).
"""

from __future__ import annotations


class AttrDict(dict):
    """A dict whose items are also attributes, converting nested dicts."""

    def __init__(self, d=None, **kwargs):
        super().__init__()
        if d is None:
            d = {}
        if kwargs:
            d = {**d, **kwargs}
        for k, v in d.items():
            self[k] = v

    @staticmethod
    def _convert(value):
        if isinstance(value, dict) and not isinstance(value, AttrDict):
            return AttrDict(value)
        if isinstance(value, (list, tuple)):
            t = type(value)
            return t(AttrDict._convert(v) for v in value)
        return value

    def __setitem__(self, key, value):
        super().__setitem__(key, AttrDict._convert(value))

    def __setattr__(self, name, value):
        self[name] = value

    def __getattr__(self, name):
        try:
            return self[name]
        except KeyError:
            raise AttributeError(name) from None

    def __delattr__(self, name):
        try:
            del self[name]
        except KeyError:
            raise AttributeError(name) from None

    def get_path(self, dotted: str, default=None):
        """``cfg.get_path("train.mmd.sigma")`` — for optional nested keys."""
        node = self
        for part in dotted.split("."):
            if isinstance(node, dict) and part in node:
                node = node[part]
            else:
                return default
        return node

    def to_dict(self) -> dict:
        def undo(v):
            if isinstance(v, AttrDict):
                return {k: undo(x) for k, x in v.items()}
            if isinstance(v, (list, tuple)):
                return type(v)(undo(x) for x in v)
            return v

        return undo(self)
