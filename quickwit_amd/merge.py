"""Cross-rank merge of LeafSearchResponse protobufs.

This is the root-side reduce of the reference (merge_fruits,
quickwit-search/src/collector.rs:832-861 + intermediate aggregation merge)
applied across GPUs: each rank leaf-searches its split batch, the packed
responses are exchanged (RCCL/gloo via torch.distributed in bench.py), and
rank 0 merges with identical tie-break semantics through the product C-ABI
(qw_merge_leaf_responses — ctx-less, CPU-side, no GPU needed)."""
import ctypes
import os

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
_lib = None


class _Buf(ctypes.Structure):
    _fields_ = [("data", ctypes.POINTER(ctypes.c_uint8)), ("len", ctypes.c_size_t)]


def _get_lib():
    global _lib
    if _lib is None:
        _lib = ctypes.CDLL(os.path.join(REPO_ROOT, "libquickwit_amd.so"))
        _lib.qw_merge_leaf_responses.argtypes = [
            ctypes.c_char_p, ctypes.c_size_t,
            ctypes.POINTER(ctypes.c_char_p), ctypes.POINTER(ctypes.c_size_t),
            ctypes.c_size_t, ctypes.POINTER(_Buf)]
        _lib.qw_buf_free.argtypes = [ctypes.POINTER(_Buf)]
    return _lib


def merge_leaf_responses(search_request_pb: bytes, response_pbs: list) -> bytes:
    lib = _get_lib()
    arr = (ctypes.c_char_p * len(response_pbs))(*response_pbs)
    lens = (ctypes.c_size_t * len(response_pbs))(*[len(r) for r in response_pbs])
    buf = _Buf()
    rc = lib.qw_merge_leaf_responses(search_request_pb, len(search_request_pb),
                                     arr, lens, len(response_pbs), ctypes.byref(buf))
    if rc != 0:
        raise RuntimeError(f"qw_merge_leaf_responses failed ({rc})")
    out = ctypes.string_at(buf.data, buf.len)
    lib.qw_buf_free(ctypes.byref(buf))
    return out


# --------------------------------------------------------------------------
# Cross-rank packed exchange (replaces all_gather_object):
#   1. fixed-size 32 B hit records (sort keys + ids) — tensor allgather
#      (RCCL over xGMI on GPU, gloo on CPU tests)
#   2. dense histogram bucket arrays — tensor SUM/MIN/MAX all_reduce
#   3. everything sparse/opaque (terms tables, sketches, failed splits,
#      resource stats) — byte-tensor allgather ("sideband"), merged on rank 0
#      through the same C-ABI merge (qw_merge_leaf_responses /
#      merge_fruits semantics, collector.rs:832-861)
# No pickled python objects cross ranks on the data path.

_SV_TAGS = ["u64", "i64", "f64", "boolean"]  # SortByValue oneof -> tag-1

DENSE_BUCKET_CAP = 1 << 22  # fallback to sideband beyond this span


def _sv_bits(sv):
    """SortByValue dict -> (tag u8, raw u64 bits)."""
    if not sv:
        return 0, 0
    for i, k in enumerate(_SV_TAGS):
        if k in sv:
            v = sv[k]
            if k == "f64":
                import struct
                return i + 1, struct.unpack("<Q", struct.pack("<d", v))[0]
            if k == "boolean":
                return i + 1, int(bool(v))
            return i + 1, v & 0xFFFFFFFFFFFFFFFF
    return 0, 0


def _sv_undo(tag, bits):
    if tag == 0:
        return None
    k = _SV_TAGS[tag - 1]
    if k == "f64":
        import struct
        return {k: struct.unpack("<d", struct.pack("<Q", bits))[0]}
    if k == "boolean":
        return {k: bool(bits)}
    if k == "i64":
        return {k: bits - (1 << 64) if bits >= (1 << 63) else bits}
    return {k: bits}


def _pack_hits(hits, split_ord, nmax):
    """-> uint8 ndarray of nmax 32 B records (zero-padded)."""
    import numpy as np
    rec = np.zeros(nmax, dtype=np.dtype(
        [("sv", "<u8"), ("sv2", "<u8"), ("doc", "<u4"), ("seg", "<u4"),
         ("split", "<u4"), ("flags", "<u2"), ("pad", "<u2")]))
    for i, h in enumerate(hits):
        t1, b1 = _sv_bits(h.get("sort_value"))
        t2, b2 = _sv_bits(h.get("sort_value2"))
        rec[i] = (b1, b2, h.get("doc_id", 0), h.get("segment_ord", 0),
                  split_ord[h.get("split_id", "")], t1 | (t2 << 3), 0)
    return rec.view(np.uint8).reshape(-1)


def _unpack_hits(buf, n, split_ids):
    import numpy as np
    rec = np.frombuffer(bytes(buf), dtype=np.dtype(
        [("sv", "<u8"), ("sv2", "<u8"), ("doc", "<u4"), ("seg", "<u4"),
         ("split", "<u4"), ("flags", "<u2"), ("pad", "<u2")]))[:n]
    hits = []
    for r in rec:
        h = {"split_id": split_ids[int(r["split"])], "doc_id": int(r["doc"]),
             "segment_ord": int(r["seg"])}
        sv = _sv_undo(int(r["flags"]) & 7, int(r["sv"]))
        sv2 = _sv_undo((int(r["flags"]) >> 3) & 7, int(r["sv2"]))
        if sv is not None:
            h["sort_value"] = sv
        if sv2 is not None:
            h["sort_value2"] = sv2
        hits.append(h)
    return hits


def _round_idx(key, interval, offset):
    i = round((key - offset) / interval)
    if abs(i * interval + offset - key) > 1e-6 * max(abs(key), 1.0):
        return None  # key off-grid: C++ parsed the request differently
    return int(i)


def distributed_merge(sreq_pb: bytes, resp_pb: bytes, split_ids: list,
                      device=None) -> bytes:
    """Merge this rank's LeafSearchResponse with every other rank's over
    torch.distributed (default group, world>1), returning the merged pb on
    rank 0 and b"" elsewhere. split_ids: this rank's split id list (each
    packed hit references an ordinal into the gathered table)."""
    import json as _json
    import struct as _struct

    import numpy as np
    import torch
    import torch.distributed as dist

    from . import proto, qagg

    world = dist.get_world_size()
    rank = dist.get_rank()
    if device is None:
        device = (torch.device("cuda", torch.cuda.current_device())
                  if torch.cuda.is_available() else torch.device("cpu"))

    resp = proto.decode("LeafSearchResponse", resp_pb)
    hits = resp.pop("partial_hits", [])
    blob = resp.get("intermediate_aggregation_result", b"")

    # ---- split the agg blob: dense-reducible histograms vs opaque residual
    sreq = proto.decode("SearchRequest", sreq_pb)
    params = {}
    if sreq.get("aggregation_request"):
        try:
            params = qagg.dense_params(_json.loads(sreq["aggregation_request"]))
        except (ValueError, KeyError):
            params = {}
    entries = qagg.parse_blob(blob) if blob else []
    dense = []  # (orig_index, entry, interval, offset)
    for i, e in enumerate(entries):
        if e.dense_eligible and e.name in params:
            dense.append([i, e, params[e.name][0], params[e.name][1]])
        else:
            e.buckets = None  # opaque residual
    # Structure agreement BEFORE the variable-shaped meta reduce: the dense
    # list is derived from each rank's OWN blob, so a rank whose split
    # failed (empty blob) would build a different-shaped tensor and either
    # deadlock or crash the collective. One fixed-shape reduce detects any
    # disagreement (count + order/sub-structure hash, min+max of both);
    # on mismatch every rank falls back to the sideband path.
    if params and dist.get_world_size() > 1:
        sig = 0
        for i, e, _iv, _ofs in dense:
            sig = (sig * 1000003 + i * 131 + len(e.sub_kinds) * 7
                   + 1) & 0x7FFFFFFF
        t_sig = torch.tensor([len(dense), sig, -len(dense), -sig],
                             dtype=torch.int64, device=device)
        dist.all_reduce(t_sig, op=dist.ReduceOp.MAX)
        mx_n, mx_s, mn_n, mn_s = (int(x) for x in t_sig.cpu())
        if mx_n != -mn_n or mx_s != -mn_s:
            for _i, e, _iv, _ofs in dense:
                e.buckets = None  # opaque residual: rides the sideband
            dense = []
    # per-agg index ranges; off-grid keys disqualify the agg (structural
    # eligibility is identical on every rank, so the fallback must be too:
    # decided from the request+blob grid check, then AND-reduced)
    meta = np.zeros(2 * len(dense) + 1, dtype=np.int64)
    BIG = 1 << 60
    ok = 1
    for di, (i, e, iv, ofs) in enumerate(dense):
        lo, hi = BIG, -BIG
        for key, dc, _s in e.buckets:
            idx = _round_idx(key, iv, ofs)
            if idx is None:
                ok = 0
                break
            lo, hi = min(lo, idx), max(hi, idx)
        meta[2 * di] = -lo
        meta[2 * di + 1] = hi
    meta[-1] = -ok
    tmeta = torch.from_numpy(meta).to(device)
    if dense:
        dist.all_reduce(tmeta, op=dist.ReduceOp.MAX)
    meta = tmeta.cpu().numpy()
    if meta[-1] == 0 and dense:  # some rank saw an off-grid key
        for _i, e, _iv, _ofs in dense:
            e.buckets = None
        dense = []

    spans = []
    for di, (i, e, iv, ofs) in enumerate(dense):
        lo, hi = -int(meta[2 * di]), int(meta[2 * di + 1])
        spans.append((lo, hi) if hi >= lo else (0, -1))
    if sum(max(0, hi - lo + 1) for lo, hi in spans) > DENSE_BUCKET_CAP:
        for _i, e, _iv, _ofs in dense:
            e.buckets = None
        dense, spans = [], []

    residual_entries = [e for e in entries
                        if not any(e is d[1] for d in dense)]
    if blob:
        resp["intermediate_aggregation_result"] = qagg.serialize_blob(
            residual_entries)

    # ---- sideband: response-minus-hits-minus-dense + split-id table
    side_pb = proto.encode("LeafSearchResponse", resp)
    ids_b = "\0".join(split_ids).encode()
    side = _struct.pack("<II", len(side_pb), len(ids_b)) + side_pb + ids_b

    # ---- sizes (one MAX all_reduce), then fixed-size allgathers
    sz = torch.tensor([len(side), len(hits)], dtype=torch.int64,
                      device=device)
    dist.all_reduce(sz, op=dist.ReduceOp.MAX)
    max_side, max_hits = int(sz[0]), int(sz[1])

    split_ord = {s: i for i, s in enumerate(split_ids)}
    hbuf = _pack_hits(hits, split_ord, max(max_hits, 1))
    hdr = np.zeros(16, dtype=np.uint8)
    hdr[:8] = np.frombuffer(_struct.pack("<II", len(side), len(hits)),
                            dtype=np.uint8)
    pay = np.concatenate([
        hdr, np.frombuffer(side, dtype=np.uint8).copy(),
        np.zeros(max_side - len(side), dtype=np.uint8), hbuf])
    t = torch.from_numpy(pay).to(device)
    gath = [torch.empty_like(t) for _ in range(world)]
    dist.all_gather(gath, t)

    # ---- dense reduce: i64 doc_count+sub_count SUM; f64 sum/sum_sq SUM,
    # min MIN, max MAX
    red = None
    if dense:
        tot = sum(hi - lo + 1 for lo, hi in spans)
        nsub = [len(d[1].sub_kinds) for d in dense]
        d_cnt = np.zeros(sum((hi - lo + 1) * (1 + ns)
                             for (lo, hi), ns in zip(spans, nsub)),
                         dtype=np.int64)
        d_sum = np.zeros(2 * sum((hi - lo + 1) * ns
                                 for (lo, hi), ns in zip(spans, nsub)),
                         dtype=np.float64)
        d_min = np.full(d_sum.size // 2, np.inf)
        d_max = np.full(d_sum.size // 2, -np.inf)
        c0 = s0 = 0
        for (lo, hi), (i, e, iv, ofs), ns in zip(spans, dense, nsub):
            n = hi - lo + 1
            for key, dc, subs in e.buckets:
                b = _round_idx(key, iv, ofs) - lo
                d_cnt[c0 + b] = dc
                for s, sp in enumerate(subs):
                    d_cnt[c0 + n * (1 + s) + b] = sp[0]
                    d_sum[s0 + 2 * (n * s + b)] = sp[1]
                    d_sum[s0 + 2 * (n * s + b) + 1] = sp[4]
                    d_min[s0 // 2 + n * s + b] = sp[2]
                    d_max[s0 // 2 + n * s + b] = sp[3]
            c0 += n * (1 + ns)
            s0 += 2 * n * ns
        tc = torch.from_numpy(d_cnt).to(device)
        dist.all_reduce(tc, op=dist.ReduceOp.SUM)
        red = [tc.cpu().numpy(), None, None, None]
        if d_sum.size:
            ts = torch.from_numpy(d_sum).to(device)
            tmn = torch.from_numpy(d_min).to(device)
            tmx = torch.from_numpy(d_max).to(device)
            dist.all_reduce(ts, op=dist.ReduceOp.SUM)
            dist.all_reduce(tmn, op=dist.ReduceOp.MIN)
            dist.all_reduce(tmx, op=dist.ReduceOp.MAX)
            red[1:] = [ts.cpu().numpy(), tmn.cpu().numpy(), tmx.cpu().numpy()]

    if rank != 0:
        return b""

    # ---- rank 0: rebuild per-rank responses, C-ABI merge, overlay dense
    per_rank = []
    for g in gath:
        b = g.cpu().numpy().tobytes()
        _slen, nh = _struct.unpack_from("<II", b, 0)
        pl, il = _struct.unpack_from("<II", b, 16)  # sideband: u32 u32 pb ids
        pb2 = b[24:24 + pl]
        ids2 = b[24 + pl:24 + pl + il].decode().split("\0") if il else []
        r = proto.decode("LeafSearchResponse", pb2)
        hb = b[16 + max_side:16 + max_side + max(max_hits, 1) * 32]
        r["partial_hits"] = _unpack_hits(hb, nh, ids2)
        per_rank.append(proto.encode("LeafSearchResponse", r))
    merged_pb = merge_leaf_responses(sreq_pb, per_rank)

    if dense:
        merged = proto.decode("LeafSearchResponse", merged_pb)
        res_entries = qagg.parse_blob(
            merged.get("intermediate_aggregation_result", b""))
        # re-insert reduced dense aggs at their original positions
        out_entries = []
        ri = iter(res_entries)
        dmap = {d[0]: d for d in dense}
        c0 = s0 = 0
        for i in range(len(entries)):
            if i in dmap:
                _, e, iv, ofs = dmap[i]
                lo, hi = spans[[d[0] for d in dense].index(i)]
                n = hi - lo + 1
                ns = len(e.sub_kinds)
                cnt, ssum, smin, smax = red
                buckets = []
                for b in range(n):
                    dc = int(cnt[c0 + b])
                    if dc == 0:
                        continue
                    subs = []
                    for s in range(ns):
                        subs.append((
                            int(cnt[c0 + n * (1 + s) + b]),
                            float(ssum[s0 + 2 * (n * s + b)]),
                            float(smin[s0 // 2 + n * s + b]),
                            float(smax[s0 // 2 + n * s + b]),
                            float(ssum[s0 + 2 * (n * s + b) + 1])))
                    buckets.append((float(lo + b) * iv + ofs, dc, subs))
                e.buckets = buckets
                out_entries.append(e)
                c0 += n * (1 + ns)
                s0 += 2 * n * ns
            else:
                out_entries.append(next(ri))
        merged["intermediate_aggregation_result"] = qagg.serialize_blob(
            out_entries)
        merged_pb = proto.encode("LeafSearchResponse", merged)
    return merged_pb
