"""QAGG1 v4 intermediate-aggregation blob codec (python side).

Wire layout is owned by csrc/qagg_format.h (encode()/decode() there are the
authority; this module restates it for the cross-rank dense-bucket reduce in
merge.py). Parse is STRUCTURAL over every agg kind so unknown/sparse aggs
round-trip as opaque bytes; only histogram kinds (1=date_histogram,
2=histogram) with stats-only sub-aggs are decomposed into buckets for the
RCCL sum-reduce path (dense arrays reduce, sparse tables
allgather).

Layout (little-endian), per qagg_format.h encode():
  u32 magic 'QAG1', u16 version=4, u16 n_aggs
  per agg: u16 name_len + name, u8 kind, u16 n_sub,
           per sub: u16 len + name + u8 sub_kind (0=stats 1=sketch)
  kind 6 (percentiles): u64 zero, u32 ne, ne x (i32 k, u64 count)
  kind 5 (metric):      5 x 8B (count,sum,min,max,sum_sq)
  kind 3 (terms):       u8 key_kind, u64 matched, f64 error_bound, u32 ne,
                        per entry: u16 klen + key + u64 count + n_sub x 40B
  else (1/2/4 buckets): u32 nb, per bucket: f64 key + u64 doc_count +
                        per sub (sketch if sub_kind==1 else 40B stats)
"""
import struct

MAGIC = 0x31474151
VERSION = 4
STATS_FMT = "<Qdddd"  # count, sum, min, max, sum_sq
STATS_SIZE = 40


class AggEntry:
    __slots__ = ("name", "kind", "sub_kinds", "raw", "body_off", "buckets")

    def __init__(self):
        self.buckets = None  # [(key, doc_count, [stats 5-tuple per sub])]

    @property
    def dense_eligible(self):
        return self.kind in (1, 2) and all(k == 0 for k in self.sub_kinds)


def parse_blob(blob: bytes):
    """-> list[AggEntry]; entries keep .raw for opaque reassembly and
    .buckets parsed when dense_eligible."""
    if not blob:
        return []
    magic, ver, n = struct.unpack_from("<IHH", blob, 0)
    if magic != MAGIC or ver != VERSION:
        raise ValueError(f"QAGG1: bad header {magic:#x} v{ver}")
    off = 8
    out = []
    for _ in range(n):
        start = off
        e = AggEntry()
        (nl,) = struct.unpack_from("<H", blob, off)
        off += 2
        e.name = blob[off:off + nl].decode()
        off += nl
        e.kind = blob[off]
        off += 1
        (ns,) = struct.unpack_from("<H", blob, off)
        off += 2
        e.sub_kinds = []
        for _s in range(ns):
            (sl,) = struct.unpack_from("<H", blob, off)
            off += 2 + sl
            e.sub_kinds.append(blob[off])
            off += 1
        e.body_off = off - start
        if e.kind == 6:
            (ne,) = struct.unpack_from("<I", blob, off + 8)
            off += 12 + ne * 12
        elif e.kind == 5:
            off += STATS_SIZE
        elif e.kind == 3:
            (ne,) = struct.unpack_from("<I", blob, off + 17)
            off += 21
            for _i in range(ne):
                (kl,) = struct.unpack_from("<H", blob, off)
                off += 2 + kl + 8 + ns * STATS_SIZE
        else:
            (nb,) = struct.unpack_from("<I", blob, off)
            off += 4
            if e.dense_eligible:
                e.buckets = []
            for _b in range(nb):
                key, dc = struct.unpack_from("<dQ", blob, off)
                off += 16
                subs = []
                for sk in e.sub_kinds:
                    if sk == 1:
                        (ne2,) = struct.unpack_from("<I", blob, off + 8)
                        off += 12 + ne2 * 12
                    else:
                        subs.append(struct.unpack_from(STATS_FMT, blob, off))
                        off += STATS_SIZE
                if e.buckets is not None:
                    e.buckets.append((key, dc, subs))
        e.raw = blob[start:off]
        out.append(e)
    if off != len(blob):
        raise ValueError("QAGG1: trailing bytes")
    return out


def serialize_blob(entries) -> bytes:
    """Inverse of parse_blob. Entries whose .buckets is set are re-encoded
    from the bucket list (reduced values); others emit .raw verbatim."""
    out = bytearray(struct.pack("<IHH", MAGIC, VERSION, len(entries)))
    for e in entries:
        if e.buckets is None:
            out += e.raw
            continue
        out += e.raw[:e.body_off]
        out += struct.pack("<I", len(e.buckets))
        for key, dc, subs in e.buckets:
            out += struct.pack("<dQ", key, dc)
            for sp in subs:
                out += struct.pack(STATS_FMT, *sp)
    return bytes(out)


def parse_interval_ms(s) -> float:
    """date_histogram fixed_interval string -> milliseconds (the subset the
    product accepts: ms/s/m/h/d suffixes)."""
    s = str(s).strip()
    for suf, mult in (("ms", 1.0), ("s", 1000.0), ("m", 60_000.0),
                      ("h", 3_600_000.0), ("d", 86_400_000.0)):
        if s.endswith(suf) and s[:-len(suf)].replace(".", "", 1).isdigit():
            return float(s[:-len(suf)]) * mult
    raise ValueError(f"unsupported interval {s!r}")


def dense_params(agg_request_json: dict):
    """{agg_name: (interval, offset)} for top-level date_histogram/histogram
    aggregations — the stride the cross-rank dense reduce aligns on."""
    out = {}
    for name, spec in (agg_request_json or {}).items():
        if not isinstance(spec, dict):
            continue
        if "date_histogram" in spec:
            dh = spec["date_histogram"]
            try:
                iv = parse_interval_ms(dh.get("fixed_interval"))
            except (ValueError, TypeError):
                continue
            out[name] = (iv, float(dh.get("offset", 0) or 0))
        elif "histogram" in spec:
            h = spec["histogram"]
            if "interval" in h:
                out[name] = (float(h["interval"]), float(h.get("offset", 0)))
    return out
