"""Small shared utilities."""

from __future__ import annotations


def stable_seed(*parts) -> int:
    """Deterministic seed from strings/ints — identical across processes
    (python's hash() is salted per process and must not be used)."""
    h = 2166136261
    for p in parts:
        for ch in str(p):
            h = ((h ^ ord(ch)) * 16777619) & 0xFFFFFFFF
        h = (h * 31 + 17) & 0xFFFFFFFF
    return h & 0x7FFFFFFF
