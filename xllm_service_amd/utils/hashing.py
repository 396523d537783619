"""Chained XXH3-128 block hashing for prefix-cache identity.

Block i's hash = XXH3-128(prev_digest || tokens of block i), so a block hash
pins the ENTIRE prefix — the same scheme as the reference's chained block
hash (reference: xllm_service/common/hash_util.cpp:18-45), on the real
xxHash implementation.
"""
from __future__ import annotations

import struct
from typing import Iterable, List, Optional

import xxhash

_SEED = 0x58_4C_4C_4D  # "XLLM"


def chain_block_hashes(token_ids: Iterable[int], block_size: int,
                       prev_digest: Optional[bytes] = None) -> List[bytes]:
    """Hashes of each FULL block of token_ids (partial tail ignored)."""
    toks = list(token_ids)
    out: List[bytes] = []
    prev = prev_digest or b""
    for i in range(0, len(toks) - block_size + 1, block_size):
        h = xxhash.xxh3_128(seed=_SEED)
        h.update(prev)
        h.update(struct.pack(f"<{block_size}i", *toks[i:i + block_size]))
        prev = h.digest()
        out.append(prev)
    return out
