"""Fieldnorm (per-doc token count) 1-byte quantization.

Restates the Lucene SmallFloat int4 scheme (3-bit mantissa + implicit leading
bit + exponent) that tantivy's fieldnorm code follows (tantivy 0.27 @
86641f72df7f79867486aa4645b66767a80cc4e8 is a non-vendored dependency of the
reference — SURVEY.md §8c). Identity for lengths 0..15; floor-rounded above.
Pinned by the reference's BM25 golden vectors (tests/golden/bm25_sort.json,
doc lengths 1..2); bit-level behavior beyond the golden range is parity
unpinned. Must stay in lockstep with
quickwit_amd/csrc/fieldnorm.h and oracle/oracle.cpp.
"""
import numpy as np


def _decode(norm_id: int) -> int:
    if norm_id < 8:
        return norm_id
    mantissa = (norm_id & 7) | 8
    exponent = (norm_id >> 3) - 1
    return mantissa << exponent


FIELDNORM_TABLE = np.array([_decode(i) for i in range(256)], dtype=np.uint64)


def norm_to_id(lengths) -> np.ndarray:
    """Encode token counts to fieldnorm ids (floor: largest id whose decoded
    value is <= length)."""
    lengths = np.asarray(lengths, dtype=np.uint64)
    ids = np.searchsorted(FIELDNORM_TABLE, lengths, side="right") - 1
    return ids.astype(np.uint8)


def id_to_norm(ids) -> np.ndarray:
    return FIELDNORM_TABLE[np.asarray(ids, dtype=np.uint8)]
