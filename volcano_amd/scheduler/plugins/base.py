from __future__ import annotations

from typing import Callable, Dict


class Plugin:
    """framework.Plugin analog: Name / OnSessionOpen / OnSessionClose."""

    name = "plugin"

    def __init__(self, args: dict):
        self.args = args or {}

    def on_session_open(self, ssn) -> None:  # pragma: no cover - interface
        pass

    def on_session_close(self, ssn) -> None:
        pass


PLUGIN_REGISTRY: Dict[str, Callable[[dict], Plugin]] = {}


def register(name: str):
    def deco(cls):
        cls.name = name
        PLUGIN_REGISTRY[name] = cls
        return cls
    return deco
