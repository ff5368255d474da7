"""Counter-based RNG — bit-exact torch mirror of hip_common.h's rng_u32.

Two murmur3-finalizer rounds over (seed ^ ctr*2654435761). All arithmetic is
uint32; torch has no uint32 mul, so low-32 products are formed from 16-bit
limbs in int64 (no overflow: each partial < 2^48).
"""

from __future__ import annotations

import torch

_M1 = 0x85EBCA6B
_M2 = 0xC2B2AE35
_PHI = 2654435761  # 2^32 / golden ratio


def _mul_lo32(a: torch.Tensor, b: int) -> torch.Tensor:
    lo = (a & 0xFFFF) * b
    hi = (((a >> 16) * b) & 0xFFFF) << 16
    return (lo + hi) & 0xFFFFFFFF


def rng_u32(seed: int, ctr: torch.Tensor) -> torch.Tensor:
    """ctr: int64 tensor of counters -> uint32 hashes (as int64)."""
    h = (seed ^ _mul_lo32(ctr & 0xFFFFFFFF, _PHI)) & 0xFFFFFFFF
    for _ in range(2):
        h = h ^ (h >> 16)
        h = _mul_lo32(h, _M1)
        h = h ^ (h >> 13)
        h = _mul_lo32(h, _M2)
        h = h ^ (h >> 16)
    return h


def rng_uniform(seed: int, ctr: torch.Tensor) -> torch.Tensor:
    """float32 uniforms in (0,1), matching the device rng_uniform."""
    return ((rng_u32(seed, ctr).to(torch.float64) + 0.5) *
            2.3283064365386963e-10).to(torch.float32)
