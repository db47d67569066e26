"""QWA1 split container writer + synthetic corpus generator.

Format spec: DESIGN.md §3. This is indexing-side tooling (the reference's
counterpart is the quickwit-indexing pipeline + tantivy's segment writer,
which are out of scope as services — SURVEY.md §2): the build needs only a
deterministic split writer for tests and benchmarks.

Layout is GPU-first: 128-doc bitpacked posting blocks with 16-byte skip
entries, raw fixed-width fast-field columns, u8 fieldnorms, all sections
64-byte aligned so the file image can be uploaded to HBM as-is and indexed by
offset (tantivy-*shaped*: same structures the reference's hot loops walk —
SURVEY.md §8a rows "posting block decode" / "fast-field predicate" — with our
own encodings, since tantivy's are unpinned).
"""
import datetime as _dt
import json

import numpy as np

from .fieldnorm import norm_to_id

MAGIC = b"QWAMDSP1"
FOOTER_MAGIC = b"QWA1FOOT"
ALIGN = 64
BLOCK = 128

SKIP_DTYPE = np.dtype(
    [
        ("first_doc", "<u4"),
        ("last_doc", "<u4"),
        ("word_off", "<u4"),  # u32-word offset into the field's payload section
        ("id_bits", "u1"),
        ("tf_bits", "u1"),  # 0 = record:basic (no tf section)
        ("count", "<u2"),
    ]
)
assert SKIP_DTYPE.itemsize == 16


# ---------------------------------------------------------------- tokenizers
def tokenize(text: str, tokenizer: str):
    """'raw' = whole string, verbatim (tantivy raw tokenizer). 'default' =
    alphanumeric runs, lowercased, tokens longer than 40 UTF-8 bytes
    dropped (tantivy default = SimpleTokenizer + RemoveLongFilter(40) +
    LowerCaser). Alnum/lowercase come from the GENERATED shared tables
    (unicode_tables.py == csrc/qw_unicode.h) so the C++ query-side
    tokenizer agrees with this writer on every codepoint."""
    from .unicode_tables import is_alnum_cp, lower_cp
    if text is None:
        return []
    if tokenizer == "raw":
        return [text] if text else []
    toks, cur = [], []
    for ch in text:
        cp = ord(ch)
        if cp < 128:
            if ("0" <= ch <= "9") or ("a" <= ch <= "z"):
                cur.append(ch)
                continue
            if "A" <= ch <= "Z":
                cur.append(chr(cp + 32))
                continue
        elif is_alnum_cp(cp):
            cur.append(chr(lower_cp(cp)))
            continue
        if cur:
            toks.append("".join(cur))
            cur = []
    if cur:
        toks.append("".join(cur))
    return [t for t in toks if len(t.encode("utf-8")) <= 40]


# ------------------------------------------------------------- bit packing
def _bit_width(vals: np.ndarray) -> np.ndarray:
    """Per-row max bit width (>=1) of a (nblk, BLOCK) uint64 matrix."""
    m = vals.max(axis=1)
    bits = np.ones(len(m), dtype=np.uint8)
    for k in range(1, 33):
        bits += (m >= (np.uint64(1) << np.uint64(k))).astype(np.uint8)
    return bits


def _pack_group(vals: np.ndarray, w: int) -> np.ndarray:
    """Pack (nb, BLOCK) uint64 values of width w into (nb, ceil(BLOCK*w/64))
    u64 words, LSB-first."""
    nb = vals.shape[0]
    nw = (BLOCK * w + 63) // 64
    packed = np.zeros((nb, nw), dtype=np.uint64)
    for j in range(BLOCK):
        p = j * w
        wd, sh = p >> 6, p & 63
        packed[:, wd] |= vals[:, j] << np.uint64(sh)
        if sh + w > 64:
            packed[:, wd + 1] |= vals[:, j] >> np.uint64(64 - sh)
    return packed


def _pack_blocks(vals: np.ndarray, widths: np.ndarray, words64_off: np.ndarray, out: np.ndarray):
    """Scatter-pack each block i (vals[i], width widths[i]) into
    out[words64_off[i] : words64_off[i] + nwords64(widths[i])]."""
    for w in np.unique(widths):
        sel = np.nonzero(widths == w)[0]
        packed = _pack_group(vals[sel], int(w))
        nw = packed.shape[1]
        idx = (words64_off[sel][:, None] + np.arange(nw, dtype=np.int64)[None, :]).ravel()
        out[idx] = packed.ravel()


# ------------------------------------------------------------- the writer
class _Sections:
    def __init__(self):
        self.chunks = [MAGIC, b"\0" * (ALIGN - len(MAGIC))]
        self.pos = ALIGN

    def add(self, data) -> list:
        b = data.tobytes() if isinstance(data, np.ndarray) else bytes(data)
        off = self.pos
        self.chunks.append(b)
        self.pos += len(b)
        pad = (-self.pos) % ALIGN
        if pad:
            self.chunks.append(b"\0" * pad)
            self.pos += pad
        return [off, len(b)]


def _build_text_field(sec, fspec, num_docs, term_ids, doc_ids, tfs, vocab, doc_lengths,
                      positions=None):
    """term_ids/doc_ids/tfs: postings sorted by (term, doc). vocab: list of
    term strings ordered by term id == sorted order. positions: u32 array
    ordered (term, doc, occurrence) when record == "position" (phrase
    queries). Returns meta dict."""
    record = fspec.get("record", "basic")
    record_freq = record in ("freq", "position")
    nterms = len(vocab)
    df = np.bincount(term_ids, minlength=nterms).astype(np.int64)
    term_start = np.concatenate([[0], np.cumsum(df)])
    nblk_t = (df + BLOCK - 1) // BLOCK
    total_blocks = int(nblk_t.sum())
    npost = len(doc_ids)

    if total_blocks:
        blk_term = np.repeat(np.arange(nterms), nblk_t)
        blk_local = np.arange(total_blocks) - np.repeat(
            np.concatenate([[0], np.cumsum(nblk_t)[:-1]]), nblk_t
        )
        blk_start = term_start[blk_term] + blk_local * BLOCK
        blk_count = np.minimum(df[blk_term] - blk_local * BLOCK, BLOCK).astype(np.int64)
        first_doc = doc_ids[blk_start]
        last_doc = doc_ids[blk_start + blk_count - 1]

        gaps = np.diff(doc_ids.astype(np.int64), prepend=0)
        gaps[blk_start] = 0
        blk_of_p = np.repeat(np.arange(total_blocks), blk_count)
        col_of_p = np.arange(npost) - blk_start[blk_of_p]
        vals = np.zeros((total_blocks, BLOCK), dtype=np.uint64)
        vals[blk_of_p, col_of_p] = gaps.astype(np.uint64)
        id_bits = _bit_width(vals)
        id_w64 = (BLOCK * id_bits.astype(np.int64) + 63) // 64

        if record_freq:
            tvals = np.zeros((total_blocks, BLOCK), dtype=np.uint64)
            tvals[blk_of_p, col_of_p] = (tfs - 1).astype(np.uint64)
            tf_bits = _bit_width(tvals)
            tf_w64 = (BLOCK * tf_bits.astype(np.int64) + 63) // 64
        else:
            tf_bits = np.zeros(total_blocks, dtype=np.uint8)
            tf_w64 = np.zeros(total_blocks, dtype=np.int64)

        # per-block SEGMENT ANCHORS (format v2): 2 u64 words before the
        # bitpacked gaps holding u32 [pad, doc31, doc63, doc95] — the decode
        # kernel scans deltas within 32-element segments only (3 shuffle
        # steps instead of 5; the dependent scan chain is the decode
        # critical path). Readers that walk word_off (oracle, splitread)
        # are unaffected: word_off still points at the gap words.
        blk_w64 = 2 + id_w64 + tf_w64
        blk_off64 = np.concatenate([[0], np.cumsum(blk_w64)[:-1]])
        payload = np.zeros(int(blk_w64.sum()), dtype=np.uint64)
        anchor_pos = [blk_start + np.minimum(k * 32 - 1, blk_count - 1)
                      for k in (1, 2, 3)]
        a1, a2, a3 = (doc_ids[p].astype(np.uint64) for p in anchor_pos)
        payload[blk_off64] = a1 << np.uint64(32)  # [pad|doc31]
        payload[blk_off64 + 1] = a2 | (a3 << np.uint64(32))
        _pack_blocks(vals, id_bits, blk_off64 + 2, payload)
        if record_freq:
            _pack_blocks(tvals, tf_bits, blk_off64 + 2 + id_w64, payload)

        skip = np.zeros(total_blocks, dtype=SKIP_DTYPE)
        skip["first_doc"] = first_doc
        skip["last_doc"] = last_doc
        skip["word_off"] = ((blk_off64 + 2) * 2).astype(np.uint32)  # u32 words
        skip["id_bits"] = id_bits
        skip["tf_bits"] = tf_bits
        skip["count"] = blk_count.astype(np.uint16)

        term_first_blk = np.concatenate([[0], np.cumsum(nblk_t)[:-1]])
        posting_off = (blk_off64[np.minimum(term_first_blk, total_blocks - 1)] * 8).astype(
            np.uint64
        )
        posting_off[df == 0] = 0
        skip_off = (term_first_blk * 16).astype(np.uint64)
    else:
        payload = np.zeros(0, dtype=np.uint64)
        skip = np.zeros(0, dtype=SKIP_DTYPE)
        posting_off = np.zeros(nterms, dtype=np.uint64)
        skip_off = np.zeros(nterms, dtype=np.uint64)

    term_bytes = "".join(vocab).encode("utf-8")
    term_offsets = np.zeros(nterms + 1, dtype=np.uint32)
    np.cumsum([len(t.encode("utf-8")) for t in vocab], out=term_offsets[1:])

    norms = norm_to_id(doc_lengths) if fspec.get("fieldnorms", True) else None

    meta = {
        "name": fspec["name"],
        "type": "text",
        "tokenizer": fspec.get("tokenizer", "default"),
        "record": record if record_freq else "basic",
        "fieldnorms": norms is not None,
        "total_tokens": int(doc_lengths.sum()),
        "num_terms": nterms,
        "sec": {
            "term_offsets": sec.add(term_offsets),
            "term_bytes": sec.add(np.frombuffer(term_bytes, dtype=np.uint8)),
            "posting_off": sec.add(posting_off),
            "doc_freq": sec.add(df.astype(np.uint32)),
            "n_blocks": sec.add(nblk_t.astype(np.uint32)),
            "skip_off": sec.add(skip_off),
            "skip": sec.add(skip),
            "payload": sec.add(payload),
        },
    }
    if norms is not None:
        meta["sec"]["fieldnorms"] = sec.add(norms)
    if record == "position":
        assert positions is not None and len(positions) == int(tfs.sum())
        # per-block cumulative tf: posting j of block b starts at
        # pos_start[b] + sum(tf of postings before j in the block)
        cum = np.concatenate([[0], np.cumsum(tfs.astype(np.int64))])
        if total_blocks:
            pos_start = cum[blk_start].astype(np.uint32)
        else:
            pos_start = np.zeros(0, np.uint32)
        meta["sec"]["pos_start"] = sec.add(pos_start)
        meta["sec"]["positions"] = sec.add(
            np.asarray(positions, dtype=np.uint32))
    return meta


def _build_multi_str_fast(sec, fspec, num_docs, values):
    """Multi-valued str fast column: per-doc SORTED DISTINCT ord lists.
    values: list aligned to docs; each entry None | str | list[str].
    Sections: value_offsets u32[num_docs+1] + ords (+ dict)."""
    per_doc = []
    allvals = set()
    for v in values:
        if v is None:
            per_doc.append([])
        elif isinstance(v, str):
            per_doc.append([v])
            allvals.add(v)
        else:
            vs = sorted({x for x in v if isinstance(x, str)})
            per_doc.append(vs)
            allvals.update(vs)
    strs = sorted(allvals)
    dct = {s: i for i, s in enumerate(strs)}
    card = len(strs)
    width = 1 if card <= 0xFF else (2 if card <= 0xFFFF else 4)
    dt = {1: np.uint8, 2: np.uint16, 4: np.uint32}[width]
    offsets = np.zeros(num_docs + 1, dtype=np.uint32)
    np.cumsum([len(p) for p in per_doc], out=offsets[1:])
    ords = np.zeros(int(offsets[-1]), dtype=dt)
    pos = 0
    for p in per_doc:
        for s in p:
            ords[pos] = dct[s]
            pos += 1
    dict_bytes = "".join(strs).encode("utf-8")
    dict_offsets = np.zeros(card + 1, dtype=np.uint32)
    np.cumsum([len(s.encode("utf-8")) for s in strs], out=dict_offsets[1:])
    meta = {"name": fspec["name"], "type": "str", "multi": True,
            "nullable": False, "cardinality": card, "ord_width": width,
            "sec": {"value_offsets": sec.add(offsets),
                    "values": sec.add(ords),
                    "dict_offsets": sec.add(dict_offsets),
                    "dict_bytes": sec.add(np.frombuffer(dict_bytes, dtype=np.uint8))}}
    return meta


def _build_fast_field(sec, fspec, num_docs, values, present):
    """values: list aligned to docs (None where absent) or np array.
    present: bool array or None (all present)."""
    ftype = fspec["type"]
    meta = {"name": fspec["name"], "type": ftype, "nullable": False, "sec": {}}
    if present is None:
        present = np.ones(num_docs, dtype=bool)
    else:
        present = np.asarray(present, dtype=bool)
    nullable = not present.all()
    meta["nullable"] = bool(nullable)

    if ftype == "str":
        if isinstance(values, tuple) and values[0] == "ords":
            # bulk path: precomputed (ords ndarray, sorted vocab list)
            _, raw_ords, strs = values
            assert list(strs) == sorted(strs)
            card = len(strs)
            width = 1 if card <= 0xFF else (2 if card <= 0xFFFF else 4)
            ords = np.asarray(raw_ords).astype(
                {1: np.uint8, 2: np.uint16, 4: np.uint32}[width]
            )
        else:
            strs = sorted({v for v, p in zip(values, present) if p})
            dct = {s: i for i, s in enumerate(strs)}
            card = len(strs)
            width = 1 if card <= 0xFF else (2 if card <= 0xFFFF else 4)
            ords = np.zeros(num_docs, dtype={1: np.uint8, 2: np.uint16, 4: np.uint32}[width])
            for d, (v, p) in enumerate(zip(values, present)):
                if p:
                    ords[d] = dct[v]
        dict_bytes = "".join(strs).encode("utf-8")
        dict_offsets = np.zeros(card + 1, dtype=np.uint32)
        np.cumsum([len(s.encode("utf-8")) for s in strs], out=dict_offsets[1:])
        meta.update({"cardinality": card, "ord_width": width})
        meta["sec"]["values"] = sec.add(ords)
        meta["sec"]["dict_offsets"] = sec.add(dict_offsets)
        meta["sec"]["dict_bytes"] = sec.add(np.frombuffer(dict_bytes, dtype=np.uint8))
    elif ftype == "mixed":
        # dynamic field with several value types per corpus (u64/i64/f64/
        # bool): sections = f64-monotonic u64 sort keys (bool as 0/1) +
        # per-doc type tag + raw typed bits. Order across types is numeric
        # (the reference's SortValue ordering); exactness beyond 2^53 is a
        # declared approximation.
        import struct as _st

        def _f64_sortable(d):
            b = _st.unpack("<Q", _st.pack("<d", d))[0]
            return (~b) & 0xFFFFFFFFFFFFFFFF if b >> 63 else b | (1 << 63)

        keys = np.zeros(num_docs, dtype=np.uint64)
        tags = np.zeros(num_docs, dtype=np.uint8)
        raw = np.zeros(num_docs, dtype=np.uint64)
        for d, (v, p) in enumerate(zip(values, present)):
            if not p:
                continue
            if isinstance(v, bool):
                tags[d] = 3
                raw[d] = int(v)
                num = 1.0 if v else 0.0
            elif isinstance(v, int):
                if v < 0:
                    tags[d] = 1
                    raw[d] = v & 0xFFFFFFFFFFFFFFFF
                else:
                    tags[d] = 0
                    raw[d] = v
                num = float(v)
            else:
                tags[d] = 2
                raw[d] = _st.unpack("<Q", _st.pack("<d", float(v)))[0]
                num = float(v)
            keys[d] = _f64_sortable(num)
        meta["sec"]["values"] = sec.add(keys)
        meta["sec"]["tags"] = sec.add(tags)
        meta["sec"]["raw"] = sec.add(raw)
    else:
        if ftype == "bool":
            # doc-mapper bool: u64 0/1 storage; the query side maps
            # true/false literals to 1/0 (SchemaField type stays bool)
            meta["type"] = ftype = "u64"
            values = [int(bool(v)) for v in values]
        dt = (np.uint64 if ftype == "u64"
              else np.float64 if ftype == "f64" else np.int64)
        col = np.zeros(num_docs, dtype=dt)
        vv = np.asarray(
            [v if p else 0 for v, p in zip(values, present)]
            if not isinstance(values, np.ndarray)
            else np.where(present, values, 0),
            dtype=dt,
        )
        col[:] = vv
        pv = col[present]
        if ftype == "f64":
            meta["min_value"] = float(pv.min()) if len(pv) else 0.0
            meta["max_value"] = float(pv.max()) if len(pv) else 0.0
        else:
            meta["min_value"] = int(pv.min()) if len(pv) else 0
            meta["max_value"] = int(pv.max()) if len(pv) else 0
        meta["sec"]["values"] = sec.add(col)
    if nullable:
        bits = np.zeros((num_docs + 63) // 64, dtype=np.uint64)
        idx = np.nonzero(present)[0]
        np.bitwise_or.at(bits, idx >> 6, np.uint64(1) << (idx & 63).astype(np.uint64))
        meta["sec"]["nulls"] = sec.add(bits)
    return meta


def parse_datetime_ms(v) -> int:
    """Accept RFC3339 strings or unix seconds/ms ints (hdfs-logs uses
    unix_timestamp input; the agg golden corpus uses rfc3339)."""
    if isinstance(v, str):
        s = v.replace("Z", "+00:00")
        # fromisoformat (3.10) only takes 3- or 6-digit fractions: normalize
        import re
        m = re.match(r"^(.*?\.)(\d+)(.*)$", s)
        if m and len(m.group(2)) not in (3, 6):
            frac = (m.group(2) + "000000")[:6]
            s = m.group(1) + frac + m.group(3)
        d = _dt.datetime.fromisoformat(s)
        return int(d.timestamp() * 1000)
    v = int(v)
    # unix seconds unless the magnitude says millis (quickwit's
    # unix_timestamp input format guesses the unit the same way)
    return v * 1000 if abs(v) < 10**12 else v


class SplitWriter:
    """Document-at-a-time writer for small corpora (golden tests).
    Bulk columnar path: build_split_from_columns."""

    def __init__(self, schema: dict, split_id: str, store_docs: bool = True):
        self.schema = schema
        self.split_id = split_id
        self.store_docs = store_docs  # write the docstore (fetch_docs phase 2)
        self.docs = []

    def add_documents(self, docs):
        self.docs.extend(docs)

    def finalize(self) -> bytes:
        num_docs = len(self.docs)
        text_inputs = {}
        fast_inputs = {}
        for fspec in self.schema["fields"]:
            name = fspec["name"]
            if fspec["type"] == "text":
                tok = fspec.get("tokenizer", "default")
                per_doc = [tokenize_value(d.get(name), tok)
                           if d.get(name) is not None else [] for d in self.docs]
                text_inputs[name] = per_doc
                if fspec.get("fast", False):
                    lower = fspec.get("fast_normalizer") == "lowercase"
                    if fspec.get("multi"):
                        vals = []
                        for d in self.docs:
                            v = d.get(name)
                            if isinstance(v, str):
                                v = [v]
                            if isinstance(v, list):
                                # empty strings are absent from the fast
                                # column (es_compat exists golden: a doc
                                # whose only value is "" does not exist)
                                vals.append([x.lower() if lower else x
                                             for x in v
                                             if isinstance(x, str) and x])
                            else:
                                vals.append(None)
                        fast_inputs[name] = (vals, None)
                    else:
                        vals, pres = [], []
                        for d in self.docs:
                            v = d.get(name)
                            ok = isinstance(v, str) and v != ""
                            pres.append(ok)
                            vals.append((v.lower() if lower else v) if ok else "")
                        fast_inputs[name] = (vals, np.array(pres, dtype=bool))
            elif fspec.get("fast", False):
                vals, pres = [], []
                for d in self.docs:
                    v = d.get(name)
                    pres.append(v is not None)
                    if fspec["type"] == "datetime" and v is not None:
                        v = parse_datetime_ms(v)
                    vals.append(v if v is not None else 0)
                fast_inputs[name] = (vals, np.array(pres, dtype=bool))
        return _assemble(self.schema, self.split_id, num_docs, text_inputs,
                         fast_inputs, docs=self.docs if self.store_docs else None)


def _postings_from_tokens(per_doc_tokens):
    """per_doc_tokens: list of token lists -> (vocab, term_ids, doc_ids, tfs)
    sorted by (term, doc), plus doc token counts and the flat positions
    array ((term, doc, occurrence) order — token index within the doc)."""
    vocab = sorted({t for toks in per_doc_tokens for t in toks})
    tid = {t: i for i, t in enumerate(vocab)}
    occ = {}
    for d, toks in enumerate(per_doc_tokens):
        for p, t in enumerate(toks):
            occ.setdefault((tid[t], d), []).append(p)
    keys = sorted(occ)
    if keys:
        arr = np.array(keys, dtype=np.int64)
        term_ids = arr[:, 0]
        doc_ids = arr[:, 1].astype(np.uint32)
        tfs = np.array([len(occ[k]) for k in keys], dtype=np.uint32)
        positions = np.array([p for k in keys for p in occ[k]],
                             dtype=np.uint32)
    else:
        term_ids = np.zeros(0, np.int64)
        doc_ids = np.zeros(0, np.uint32)
        tfs = np.zeros(0, np.uint32)
        positions = np.zeros(0, np.uint32)
    lengths = np.array([len(t) for t in per_doc_tokens], dtype=np.int64)
    return vocab, term_ids, doc_ids, tfs, lengths, positions


DOCSTORE_BLOCK_BYTES = 16 * 1024  # uncompressed block target (tantivy-shaped)


def _build_docstore(sec, docs):
    """Row store: zlib-compressed blocks of canonical-JSON docs + block index
    (u32 first_doc[n+1], u64 block_off[n+1]). The on-disk encoding is ours
    (tantivy's docstore byte layout is unpinned); the reference
    shape is the same: compressed blocks + doc->block checkpoints."""
    import zlib
    doc_jsons = [json.dumps(d, sort_keys=True, separators=(",", ":")).encode()
                 for d in docs]
    firsts, offs, blobs = [], [], []
    pos = 0
    i = 0
    while i < len(doc_jsons):
        j = i
        size = 0
        while j < len(doc_jsons) and (j == i or size < DOCSTORE_BLOCK_BYTES):
            size += len(doc_jsons[j]) + 4
            j += 1
        block = np.array([len(b) for b in doc_jsons[i:j]], dtype="<u4").tobytes()
        block += b"".join(doc_jsons[i:j])
        comp = zlib.compress(block, 6)
        firsts.append(i)
        offs.append(pos)
        blobs.append(comp)
        pos += len(comp)
        i = j
    firsts.append(len(doc_jsons))
    offs.append(pos)
    index = (np.array(firsts, dtype="<u4").tobytes() +
             np.array(offs, dtype="<u8").tobytes())
    return {
        "n_blocks": len(blobs),
        "doc_index": sec.add(np.frombuffer(index, dtype=np.uint8)),
        "doc_blocks": sec.add(np.frombuffer(b"".join(blobs) or b"\0", dtype=np.uint8)),
    }


def _assemble(schema, split_id, num_docs, text_inputs, fast_inputs, precomputed_text=None,
              docs=None):
    sec = _Sections()
    fields_meta = []
    for fspec in schema["fields"]:
        name = fspec["name"]
        if fspec["type"] == "text":
            positions = None
            if precomputed_text and name in precomputed_text:
                vocab, term_ids, doc_ids, tfs, lengths = precomputed_text[name]
                assert fspec.get("record") != "position", \
                    "positions need the document-at-a-time writer"
            else:
                (vocab, term_ids, doc_ids, tfs, lengths,
                 positions) = _postings_from_tokens(text_inputs[name])
            fields_meta.append(
                _build_text_field(sec, fspec, num_docs, term_ids, doc_ids,
                                  tfs, vocab, lengths, positions=positions)
            )
            if fspec.get("fast", False):
                # text field with fast=true (quickwit dynamic mapping): BOTH
                # an inverted index (tokenized) and a raw-string str fast
                # column under the same name — two meta entries, found by
                # text_field()/fast_field() respectively
                vals, pres = fast_inputs[name]
                if fspec.get("multi"):
                    fmeta = _build_multi_str_fast(
                        sec, {"name": name}, num_docs, vals)
                else:
                    fmeta = _build_fast_field(
                        sec, {"name": name, "type": "str"}, num_docs, vals, pres)
                if fspec.get("fast_normalizer") == "lowercase":
                    fmeta["normalizer"] = "lowercase"
                fields_meta.append(fmeta)
        elif fspec.get("fast", False):
            vals, pres = fast_inputs[name]
            fields_meta.append(_build_fast_field(sec, fspec, num_docs, vals, pres))

    meta = {
        "format": "QWA1",
        "version": 2,
        "split_id": split_id,
        "num_docs": int(num_docs),
        "timestamp_field": schema.get("timestamp_field"),
        "fields": fields_meta,
    }
    if docs is not None:
        meta["docstore"] = _build_docstore(sec, docs)
    meta_b = json.dumps(meta, sort_keys=True).encode("utf-8")
    meta_off = sec.pos
    sec.chunks.append(meta_b)
    sec.pos += len(meta_b)
    footer = np.array([meta_off, len(meta_b)], dtype="<u8").tobytes() + FOOTER_MAGIC
    sec.chunks.append(footer)
    return b"".join(sec.chunks)


def build_split_from_columns(schema, split_id, num_docs, text_token_ids, text_vocabs, fast_columns):
    """Bulk path for synthetic corpora.

    text_token_ids[name]: int32 (num_docs, L) matrix of ids into
    text_vocabs[name] (a sorted list of token strings); -1 = empty slot.
    fast_columns[name]: (values ndarray, present ndarray or None).
    """
    precomputed = {}
    for name, mat in text_token_ids.items():
        vocab = text_vocabs[name]
        assert list(vocab) == sorted(vocab)
        n, L = mat.shape
        doc_of = np.broadcast_to(np.arange(n, dtype=np.int64)[:, None], mat.shape)
        mask = mat >= 0
        terms = mat[mask].astype(np.int64)
        docs = doc_of[mask]
        lengths = mask.sum(axis=1).astype(np.int64)
        key = terms * np.int64(num_docs) + docs
        ukey, tfs = np.unique(key, return_counts=True)
        term_ids = ukey // num_docs
        doc_ids = (ukey % num_docs).astype(np.uint32)
        precomputed[name] = (list(vocab), term_ids, doc_ids, tfs.astype(np.uint32), lengths)
    fast_inputs = {k: (v[0], v[1]) for k, v in fast_columns.items()}
    return _assemble(schema, split_id, num_docs, {}, fast_inputs, precomputed_text=precomputed)


# ------------------------------------------------------- synthetic corpus
HDFS_SCHEMA = {
    "timestamp_field": "timestamp",
    "fields": [
        {"name": "timestamp", "type": "datetime", "fast": True},
        {"name": "tenant_id", "type": "u64", "fast": True},
        {"name": "tenant_name", "type": "str", "fast": True},
        {"name": "severity_text", "type": "text", "tokenizer": "raw", "record": "basic",
         "fieldnorms": True},
        {"name": "body", "type": "text", "tokenizer": "default", "record": "freq",
         "fieldnorms": True},
    ],
}

SEVERITIES = ["DEBUG", "INFO", "WARN", "ERROR", "FATAL"]
SEVERITY_P = [0.50, 0.40, 0.07, 0.029, 0.001]
BODY_VOCAB_SIZE = 10_000
BODY_TOKENS_PER_DOC = 10
N_TENANTS = 1_000
T0_EPOCH_S = 1_700_000_000  # synthetic 30-day window start


def body_vocab():
    return ["w%05d" % i for i in range(BODY_VOCAB_SIZE)]  # already sorted


def zipf_probs(n, s=1.0):
    p = 1.0 / np.arange(1, n + 1) ** s
    return p / p.sum()


def tokenize_value(v, tok):
    if isinstance(v, list):
        out = []
        for item in v:
            if isinstance(item, str):
                out.extend(tokenize(item, tok))
        return out
    return tokenize(v, tok)


_NATIVE_GEN = None


def _native_gen():
    """ctypes handle to libqwsplitgen.so (csrc/splitgen_native.cpp), the
    OpenMP generator — same QWA1 output, ~20x faster (the numpy path takes
    minutes at 100M docs). None when the .so is absent."""
    global _NATIVE_GEN
    if _NATIVE_GEN is None:
        import ctypes
        import os
        path = os.path.join(os.path.dirname(os.path.dirname(
            os.path.abspath(__file__))), "libqwsplitgen.so")
        if not os.path.exists(path):
            _NATIVE_GEN = False
        else:
            lib = ctypes.CDLL(path)
            lib.qw_gen_split.restype = ctypes.c_int64
            lib.qw_gen_split.argtypes = [ctypes.c_int64] * 3 + [
                ctypes.POINTER(ctypes.POINTER(ctypes.c_uint8)),
                ctypes.POINTER(ctypes.c_size_t)]
            lib.qw_gen_free.argtypes = [ctypes.POINTER(ctypes.c_uint8)]
            lib.qw_gen_set_threads.argtypes = [ctypes.c_int]
            _NATIVE_GEN = lib
    return _NATIVE_GEN or None


def generate_split(split_ord: int, num_docs: int, seed: int = 42) -> bytes:
    """Seeded hdfs-logs-style synthetic split: timestamp
    uniform over 30 days, tenant_id Zipf over 1k tenants, severity_text from
    the fixed distribution, body ~ 10 tokens/doc Zipf over a 10k vocabulary.

    Uses the native OpenMP generator when built (same format + distributions,
    per-term posting synthesis — csrc/splitgen_native.cpp); falls back to the
    numpy token-matrix path below. The two paths draw from different RNGs, so
    their bytes differ; every consumer of a given split (oracle, product,
    bench accounting) reads the same generated bytes, so parity is unaffected
    by which path produced them."""
    lib = _native_gen()
    if lib is not None:
        import ctypes
        import os as _os
        nt = _os.environ.get("QW_GEN_THREADS")
        if nt:
            lib.qw_gen_set_threads(int(nt))
        p = ctypes.POINTER(ctypes.c_uint8)()
        n = ctypes.c_size_t()
        rc = lib.qw_gen_split(split_ord, num_docs, seed,
                              ctypes.byref(p), ctypes.byref(n))
        if rc != 0:
            raise RuntimeError(f"qw_gen_split failed: {rc}")
        try:
            # NOT ctypes.string_at: its size argument truncates to 32 bits
            # (a 100M-doc split is ~3.6 GB)
            view = (ctypes.c_char * n.value).from_address(
                ctypes.addressof(p.contents))
            return bytes(view)
        finally:
            lib.qw_gen_free(p)
    return generate_split_numpy(split_ord, num_docs, seed)


def generate_split_numpy(split_ord: int, num_docs: int, seed: int = 42) -> bytes:
    """numpy reference path (token-matrix model). Token ids are assigned so
    that df rank follows the Zipf draw (vocab is sorted lexically; id order ==
    lexical order by construction of w%05d)."""
    rng = np.random.default_rng(np.random.SeedSequence([seed, split_ord]))
    ts_ms = (
        T0_EPOCH_S + rng.integers(0, 30 * 86400, size=num_docs, dtype=np.int64)
    ) * 1000
    tenants = rng.choice(N_TENANTS, size=num_docs, p=zipf_probs(N_TENANTS)).astype(np.uint64)
    sev = rng.choice(len(SEVERITIES), size=num_docs, p=SEVERITY_P).astype(np.int32)
    body = rng.choice(
        BODY_VOCAB_SIZE, size=(num_docs, BODY_TOKENS_PER_DOC), p=zipf_probs(BODY_VOCAB_SIZE)
    ).astype(np.int32)

    sev_vocab = sorted(SEVERITIES)
    sev_map = np.array([sev_vocab.index(s) for s in SEVERITIES], dtype=np.int32)
    sev_ids = sev_map[sev][:, None]  # (N,1) single raw token per doc

    tenant_names = ["t%04d" % i for i in range(N_TENANTS)]
    return build_split_from_columns(
        HDFS_SCHEMA,
        f"synthetic-{seed}-{split_ord:04d}",
        num_docs,
        {"severity_text": sev_ids, "body": body},
        {"severity_text": sev_vocab, "body": body_vocab()},
        {
            "timestamp": (ts_ms, None),
            "tenant_id": (tenants, None),
            "tenant_name": (("ords", tenants, tenant_names), None),
        },
    )


def concat_segments(segment_images: list, split_id: str) -> bytes:
    """QWA2 multi-segment split container: outer magic "QWAMDSP2" +
    per-segment complete QWA1 images (64-aligned) + outer meta JSON +
    QWA1-shaped footer. The reference's splits carry multiple tantivy
    segments (collector.rs:475-594 collects per segment); readers
    (product qw_ctx_add_split, oracle, splitread.Split) run each segment
    as its own doc-id space with PartialHit.segment_ord set."""
    chunks = [b"QWAMDSP2", b"\0" * 56]
    pos = 64
    segs = []
    total = 0
    for img in segment_images:
        assert img[:8] == MAGIC and img[-8:] == FOOTER_MAGIC, "not a QWA1 image"
        moff, mlen = np.frombuffer(img[-24:-8], dtype="<u8")
        ndocs = json.loads(img[int(moff):int(moff + mlen)])["num_docs"]
        off = pos
        chunks.append(img)
        pos += len(img)
        pad = (-pos) % ALIGN
        if pad:
            chunks.append(b"\0" * pad)
            pos += pad
        segs.append({"off": int(off), "len": len(img), "num_docs": int(ndocs)})
        total += int(ndocs)
    meta = json.dumps({"format": "QWA2", "version": 1, "split_id": split_id,
                       "num_docs": total, "segments": segs},
                      sort_keys=True).encode()
    meta_off = pos
    chunks.append(meta)
    footer = np.array([meta_off, len(meta)], dtype="<u8").tobytes() + FOOTER_MAGIC
    chunks.append(footer)
    return b"".join(chunks)
