"""Process-global configuration: YAML file merged with CLI flags.

Replaces the reference's external `theconf` dependency (used at e.g.
reference train.py:326, search.py:142) with a self-contained implementation.
A config is a nested dict loaded from a YAML file (``-c/--config``) and
mutated freely afterwards; ``Config.get()`` returns the process singleton.
Ray-style worker rehydration (reference search.py:62-64) is supported via
``Config.replace(conf_dict)``.
"""
from __future__ import annotations

import argparse
import copy
from typing import Any, Dict, Optional

import yaml


class Config:
    _instance: Optional["Config"] = None

    def __init__(self, conf: Optional[Dict[str, Any]] = None):
        self.conf: Dict[str, Any] = conf if conf is not None else {}

    # -- singleton management -------------------------------------------------
    @classmethod
    def get(cls) -> "Config":
        if cls._instance is None:
            cls._instance = Config()
        return cls._instance

    @classmethod
    def load(cls, path: str) -> "Config":
        with open(path) as f:
            conf = yaml.safe_load(f) or {}
        inst = cls.get()
        inst.conf = conf
        inst.conf.setdefault("_config_path", path)
        return inst

    @classmethod
    def replace(cls, conf: Dict[str, Any]) -> "Config":
        """Install a deep copy of ``conf`` as the process config (worker rehydrate)."""
        inst = cls.get()
        inst.conf = copy.deepcopy(conf)
        return inst

    @classmethod
    def clear(cls) -> None:
        cls._instance = None

    # -- dict-like access -----------------------------------------------------
    def __getitem__(self, key: str) -> Any:
        return self.conf[key]

    def __setitem__(self, key: str, value: Any) -> None:
        self.conf[key] = value

    def __contains__(self, key: str) -> bool:
        return key in self.conf

    def get_value(self, key: str, default: Any = None) -> Any:
        return self.conf.get(key, default)

    def dump(self) -> Dict[str, Any]:
        return copy.deepcopy(self.conf)


class ConfigArgumentParser(argparse.ArgumentParser):
    """argparse parser that also loads ``-c/--config`` YAML into Config.get().

    Mirrors the CLI contract of the reference's theconf.ConfigArgumentParser:
    flags defined with ``add_argument`` override same-named YAML keys when
    passed explicitly on the command line.
    """

    def __init__(self, *args, **kwargs):
        kwargs.setdefault("conflict_handler", "resolve")
        super().__init__(*args, **kwargs)
        self.add_argument("-c", "--config", type=str, required=False, default=None)
        self._override_keys = []

    def add_override_argument(self, *names, key: str, **kwargs):
        """Register a flag that overrides a (possibly dotted) config key."""
        self.add_argument(*names, **kwargs)
        dest = names[-1].lstrip("-").replace("-", "_")
        self._override_keys.append((dest, key))

    def parse_args(self, args=None, namespace=None):
        ns, extra = super().parse_known_args(args, namespace)
        if ns.config:
            Config.load(ns.config)
        conf = Config.get()
        # theconf-style generic overrides: any leftover `--key value` (dotted
        # keys allowed) that names an EXISTING conf entry overrides it, with
        # YAML type coercion; unknown keys still error via the strict parse.
        pairs = self._apply_generic_overrides(conf, extra)
        if args is None:
            import sys
            args = list(sys.argv[1:])
        remaining = []
        pending = list(pairs)
        i = 0
        while i < len(args):
            hit = next((p for p in pending
                        if args[i] == p[0] and i + 1 < len(args) and args[i + 1] == p[1]),
                       None)
            if hit is not None:
                pending.remove(hit)
                i += 2
                continue
            remaining.append(args[i])
            i += 1
        ns = super().parse_args(remaining, namespace)
        for dest, key in self._override_keys:
            val = getattr(ns, dest, None)
            if val is not None:
                node = conf.conf
                parts = key.split(".")
                for p in parts[:-1]:
                    node = node.setdefault(p, {})
                node[parts[-1]] = val
        return ns

    @staticmethod
    def _apply_generic_overrides(conf, extra):
        pairs = []
        i = 0
        while i < len(extra):
            tok = extra[i]
            if tok.startswith("--") and i + 1 < len(extra):
                key = tok[2:].replace("-", "_")
                parts = key.split(".")
                node = conf.conf
                ok = True
                for p in parts[:-1]:
                    if isinstance(node, dict) and p in node:
                        node = node[p]
                    else:
                        ok = False
                        break
                if ok and isinstance(node, dict) and parts[-1] in node:
                    node[parts[-1]] = yaml.safe_load(extra[i + 1])
                    pairs.append((tok, extra[i + 1]))
                    i += 2
                    continue
            i += 1
        return pairs
