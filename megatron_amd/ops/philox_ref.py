"""Bit-exact numpy mirror of the in-kernel philox4x32-10 attention-dropout
mask (ops/csrc/common.h philox10_ctr64 + flash_attn.hip attn_drop_keep).

Used by the CPU reference path of flash attention with dropout and by the
GPU parity tests: same (seed, offset) -> identical keep masks, so the HIP
kernels can be checked against a plain fp32 reference even with dropout on.
"""

from __future__ import annotations

import numpy as np

_M0 = np.uint64(0xD2511F53)
_M1 = np.uint64(0xCD9E8D57)
_W0 = np.uint32(0x9E3779B9)
_W1 = np.uint32(0xBB67AE85)
_MASK32 = np.uint64(0xFFFFFFFF)


def _philox10(seed: int, offset: int, n: np.ndarray) -> np.ndarray:
    """philox4x32-10 with 64-bit counter n (uint64 array). Returns the four
    output words stacked on the last axis (uint32, shape n.shape + (4,))."""
    n = n.astype(np.uint64)
    x = (n & _MASK32).astype(np.uint32)
    y = (n >> np.uint64(32)).astype(np.uint32)
    z = np.full_like(x, np.uint32(offset & 0xFFFFFFFF))
    w = np.full_like(x, np.uint32((offset >> 32) & 0xFFFFFFFF))
    kx = np.uint32(seed & 0xFFFFFFFF)
    ky = np.uint32((seed >> 32) & 0xFFFFFFFF)
    for _ in range(10):
        p0 = _M0 * x.astype(np.uint64)
        p1 = _M1 * z.astype(np.uint64)
        hi0 = (p0 >> np.uint64(32)).astype(np.uint32)
        lo0 = (p0 & _MASK32).astype(np.uint32)
        hi1 = (p1 >> np.uint64(32)).astype(np.uint32)
        lo1 = (p1 & _MASK32).astype(np.uint32)
        x, y, z, w = hi1 ^ y ^ kx, lo1, hi0 ^ w ^ ky, lo0
        kx = np.uint32((int(kx) + int(_W0)) & 0xFFFFFFFF)
        ky = np.uint32((int(ky) + int(_W1)) & 0xFFFFFFFF)
    return np.stack([x, y, z, w], axis=-1)


def attn_dropout_mask(seed: int, offset: int, b: int, h: int, Hq: int,
                      Sq: int, Sk: int, p: float) -> np.ndarray:
    """Keep-mask (bool, [Sq, Sk]) for head (b, h) — the exact in-kernel
    mapping: e = ((b*Hq + h)*Sq + row)*Skp + col, n = e>>2, word = e&3,
    keep = ((word >> 8) * 2^-24) > p."""
    Skp = (Sk + 3) & ~3
    rows = np.arange(Sq, dtype=np.uint64)[:, None]
    cols = np.arange(Sk, dtype=np.uint64)[None, :]
    e = ((np.uint64((b * Hq + h) * Sq) + rows) * np.uint64(Skp) + cols)
    n = e >> np.uint64(2)
    comp = (e & np.uint64(3)).astype(np.int64)
    words = _philox10(seed, offset, n)
    sel = np.take_along_axis(words, comp[..., None], axis=-1)[..., 0]
    uni = ((sel >> np.uint32(8)).astype(np.float32)
           * np.float32(1.0 / 16777216.0))
    return uni > np.float32(p)
