"""Fault injection registry.

Analog of the reference's `curvine-fault` crate
(/root/reference/crates/core/curvine-fault/src/lib.rs:48-50 registered
fault points via distributed_slice; model.rs:192-242 rule matcher with
delay-or-error actions).  Code declares fault points with
``fault_point("name")``; tests install rules that delay or raise at
matching points.  Zero overhead when no rules are installed.
"""
from __future__ import annotations

import fnmatch
import threading
import time
from dataclasses import dataclass
from typing import Optional

_rules: list["FaultRule"] = []
_lock = threading.Lock()
_registered: set[str] = set()


@dataclass
class FaultRule:
    pattern: str                      # fnmatch over point names
    action: str = "error"             # error | delay
    error: Optional[Exception] = None
    delay_s: float = 0.0
    max_hits: int = -1                # -1 = unlimited
    hits: int = 0

    def matches(self, point: str) -> bool:
        return fnmatch.fnmatch(point, self.pattern)


def fault_point(name: str) -> None:
    """Declare + evaluate a fault point. Raises/sleeps per active rules."""
    _registered.add(name)
    if not _rules:
        return
    with _lock:
        for rule in _rules:
            if not rule.matches(name):
                continue
            if rule.max_hits >= 0 and rule.hits >= rule.max_hits:
                continue
            rule.hits += 1
            action, delay, exc = rule.action, rule.delay_s, rule.error
            break
        else:
            return
    if action == "delay":
        time.sleep(delay)
    else:
        raise exc if exc is not None else RuntimeError(f"fault injected at {name}")


def install(pattern: str, action: str = "error",
            error: Exception | None = None, delay_s: float = 0.0,
            max_hits: int = -1) -> FaultRule:
    rule = FaultRule(pattern, action, error, delay_s, max_hits)
    with _lock:
        _rules.append(rule)
    return rule


def remove(rule: FaultRule) -> None:
    with _lock:
        if rule in _rules:
            _rules.remove(rule)


def clear() -> None:
    with _lock:
        _rules.clear()


def registered_points() -> set[str]:
    return set(_registered)
