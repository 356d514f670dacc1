"""Process-stable seeding.

Python's built-in ``hash`` is salted per interpreter process
(PYTHONHASHSEED), so it must never feed RNG seeds in a multi-process
engine: every derived seed here goes through blake2b of the repr, which is
stable across processes, ranks and machines.
"""

from __future__ import annotations

import hashlib
from typing import Any


def stable_hash(*parts: Any) -> int:
    """Deterministic 31-bit seed from arbitrary repr-able parts."""
    h = hashlib.blake2b(repr(parts).encode(), digest_size=8)
    return int.from_bytes(h.digest(), "little") & 0x7FFFFFFF
