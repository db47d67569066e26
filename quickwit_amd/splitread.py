"""Pure-numpy QWA1 reader — test/debug tooling only.

The product reader is quickwit_amd/csrc/split.h (C++); the CPU restatement is
oracle/oracle.cpp. This module exists so writer tests can round-trip the
format without native code and so fixtures can be inspected.
"""
import json

import numpy as np

from .splitgen import BLOCK, FOOTER_MAGIC, MAGIC, SKIP_DTYPE


class Split:
    def __init__(self, data: bytes):
        assert data[:8] == MAGIC, "bad magic"
        assert data[-8:] == FOOTER_MAGIC, "bad footer"
        meta_off, meta_len = np.frombuffer(data[-24:-8], dtype="<u8")
        self.meta = json.loads(data[int(meta_off) : int(meta_off + meta_len)])
        self.data = data
        # text+fast fields appear as TWO meta entries under one name
        # (inverted index + str fast column); keep them separately
        self.fields = {}
        self.fast_meta = {}
        for f in self.meta["fields"]:
            if f["type"] == "text":
                self.fields[f["name"]] = f
            else:
                self.fast_meta[f["name"]] = f
                self.fields.setdefault(f["name"], f)

    def _sec(self, field, name, dtype):
        meta = self.fields[field]
        if name in ("values", "nulls", "dict_offsets", "dict_bytes") and \
                field in self.fast_meta:
            meta = self.fast_meta[field]
        off, ln = meta["sec"][name]
        return np.frombuffer(self.data, dtype=dtype, count=ln // np.dtype(dtype).itemsize,
                             offset=off)

    # ---- text fields
    def terms(self, field):
        offs = self._sec(field, "term_offsets", "<u4")
        blob = self._sec(field, "term_bytes", "u1").tobytes()
        return [blob[offs[i]: offs[i + 1]].decode() for i in range(len(offs) - 1)]

    def term_id(self, field, term):
        tl = self.terms(field)
        import bisect

        i = bisect.bisect_left(tl, term)
        return i if i < len(tl) and tl[i] == term else None

    def postings(self, field, term):
        """Decode the full posting list of `term` -> (doc_ids u32, tfs u32)."""
        tid = self.term_id(field, term)
        if tid is None:
            return np.zeros(0, np.uint32), np.zeros(0, np.uint32)
        f = self.fields[field]
        doc_freq = self._sec(field, "doc_freq", "<u4")[tid]
        n_blocks = self._sec(field, "n_blocks", "<u4")[tid]
        skip_off = self._sec(field, "skip_off", "<u8")[tid]
        skip_all = self._sec(field, "skip", SKIP_DTYPE)
        first_blk = int(skip_off) // 16
        payload = self._sec(field, "payload", "<u4")
        docs, tfs = [], []
        for b in range(first_blk, first_blk + int(n_blocks)):
            e = skip_all[b]
            cnt = int(e["count"])
            gaps = _unpack(payload, int(e["word_off"]), int(e["id_bits"]), cnt)
            d = int(e["first_doc"]) + np.cumsum(gaps)
            assert d[-1] == e["last_doc"], (d[-1], e["last_doc"])
            docs.append(d.astype(np.uint32))
            if e["tf_bits"] > 0:
                tf_woff = int(e["word_off"]) + 2 * ((BLOCK * int(e["id_bits"]) + 63) // 64)
                tfs.append(
                    (_unpack(payload, tf_woff, int(e["tf_bits"]), cnt) + 1).astype(np.uint32)
                )
            else:
                tfs.append(np.ones(cnt, np.uint32))
        docs = np.concatenate(docs) if docs else np.zeros(0, np.uint32)
        tfs = np.concatenate(tfs) if tfs else np.zeros(0, np.uint32)
        assert len(docs) == doc_freq
        return docs, tfs

    def fieldnorms(self, field):
        return self._sec(field, "fieldnorms", "u1")

    # ---- fast fields
    def fast_column(self, field):
        f = self.fast_meta.get(field, self.fields[field])
        if f["type"] == "str":
            w = f["ord_width"]
            vals = self._sec(field, "values", {1: "u1", 2: "<u2", 4: "<u4"}[w])
        elif f["type"] == "u64":
            vals = self._sec(field, "values", "<u8")
        elif f["type"] == "f64":
            vals = self._sec(field, "values", "<f8")
        else:  # i64 / datetime
            vals = self._sec(field, "values", "<i8")
        present = None
        if f.get("nullable"):
            bits = self._sec(field, "nulls", "<u8")
            d = np.arange(self.meta["num_docs"])
            present = (bits[d >> 6] >> (d & 63).astype(np.uint64)) & np.uint64(1) > 0
        return vals, present

    def str_dict(self, field):
        offs = self._sec(field, "dict_offsets", "<u4")
        blob = self._sec(field, "dict_bytes", "u1").tobytes()
        return [blob[offs[i]: offs[i + 1]].decode() for i in range(len(offs) - 1)]


def _unpack(payload_u32, word_off, bits, count):
    """Inverse of splitgen packing: LSB-first `bits`-wide values."""
    out = np.zeros(count, dtype=np.uint64)
    words = payload_u32[word_off:].astype(np.uint64)
    for j in range(count):
        p = j * bits
        wd, sh = p >> 5, p & 31
        v = words[wd] >> np.uint64(sh)
        got = 32 - sh
        k = 1
        while got < bits:
            v |= words[wd + k] << np.uint64(got)
            got += 32
            k += 1
        out[j] = v & ((np.uint64(1) << np.uint64(bits)) - np.uint64(1))
    return out


class MultiSplit:
    """QWA2 reader: .segments = list of Split, one per segment."""

    def __init__(self, data: bytes):
        assert data[:8] == b"QWAMDSP2", "not a QWA2 container"
        assert data[-8:] == FOOTER_MAGIC
        meta_off, meta_len = np.frombuffer(data[-24:-8], dtype="<u8")
        self.meta = json.loads(data[int(meta_off): int(meta_off + meta_len)])
        assert self.meta["format"] == "QWA2"
        self.segments = [
            Split(data[s["off"]: s["off"] + s["len"]])
            for s in self.meta["segments"]
        ]
