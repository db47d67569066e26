"""ctypes bindings for the product C-ABI (libquickwit_amd.so) and the CPU
oracle (oracle/liboracle.so — test infrastructure only).

Both expose the same driver interface so tests/bench run identical request
flows on either side:
    s = OracleSearcher() | GpuSearcher(device=0)
    s.add_split(split_id, qwa1_bytes)
    resp_dict = s.leaf_search(leaf_search_request_dict)  # proto.py dicts
"""
import ctypes
import json
import os

from . import proto

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


class _Buf(ctypes.Structure):
    _fields_ = [("data", ctypes.POINTER(ctypes.c_uint8)), ("len", ctypes.c_size_t)]


def _bind(lib, syms):
    """syms: dict role -> exported symbol name."""
    getattr(lib, syms["create"]).restype = ctypes.c_void_p
    getattr(lib, syms["add_split"]).argtypes = [
        ctypes.c_void_p, ctypes.c_char_p, ctypes.c_char_p, ctypes.c_size_t]
    getattr(lib, syms["leaf_search"]).argtypes = [
        ctypes.c_void_p, ctypes.c_char_p, ctypes.c_size_t, ctypes.POINTER(_Buf)]
    getattr(lib, syms["last_error"]).restype = ctypes.c_char_p
    getattr(lib, syms["last_error"]).argtypes = [ctypes.c_void_p]
    getattr(lib, syms["buf_free"]).argtypes = [ctypes.POINTER(_Buf)]


class _BaseSearcher:
    _syms = None
    _libname = None

    def __init__(self, create_arg=None):
        # QW_PRODUCT_LIB: perf-experiment override for the product library
        # path (e.g. a -DQW_TILE_DOCS variant built alongside the default)
        override = os.environ.get("QW_PRODUCT_LIB")
        if override and self._libname == "libquickwit_amd.so":
            path = override
        else:
            path = os.path.join(REPO_ROOT, self._libname)
        if not os.path.exists(path):
            raise FileNotFoundError(
                f"{path} not built — run __graft_entry__.build() first")
        self._lib = ctypes.CDLL(path)
        _bind(self._lib, self._syms)
        create = getattr(self._lib, self._syms["create"])
        if create_arg is None:
            self._ctx = create()
        else:
            create.argtypes = [ctypes.c_char_p]
            self._ctx = create(create_arg)
        if not self._ctx:
            raise RuntimeError("ctx creation failed")

    def _err(self):
        msg = getattr(self._lib, self._syms["last_error"])(self._ctx)
        return msg.decode() if msg else "unknown error"

    def add_split(self, split_id: str, data: bytes):
        rc = getattr(self._lib, self._syms["add_split"])(
            self._ctx, split_id.encode(), data, len(data))
        if rc != 0:
            raise RuntimeError(f"add_split failed ({rc}): {self._err()}")

    def leaf_search_raw(self, req_pb: bytes) -> bytes:
        buf = _Buf()
        rc = getattr(self._lib, self._syms["leaf_search"])(
            self._ctx, req_pb, len(req_pb), ctypes.byref(buf))
        if rc != 0:
            raise RuntimeError(f"leaf_search failed ({rc}): {self._err()}")
        out = ctypes.string_at(buf.data, buf.len)
        getattr(self._lib, self._syms["buf_free"])(ctypes.byref(buf))
        return out

    def leaf_search(self, leaf_req: dict) -> dict:
        req_pb = proto.encode("LeafSearchRequest", leaf_req)
        return proto.decode("LeafSearchResponse", self.leaf_search_raw(req_pb))


class OracleSearcher(_BaseSearcher):
    """CPU restatement (oracle) — tests/bench cpu_baseline only."""

    _syms = {"create": "qw_oracle_create", "add_split": "qw_oracle_add_split",
             "leaf_search": "qw_oracle_leaf_search",
             "last_error": "qw_oracle_last_error", "buf_free": "qw_oracle_buf_free"}
    _libname = "oracle/liboracle.so"

    def finalize_agg_json(self, blob: bytes, agg_request: dict) -> dict:
        fn = self._lib.qw_oracle_finalize_agg
        fn.argtypes = [ctypes.c_char_p, ctypes.c_size_t, ctypes.c_char_p,
                       ctypes.POINTER(_Buf)]
        buf = _Buf()
        rc = fn(blob, len(blob), json.dumps(agg_request).encode(), ctypes.byref(buf))
        if rc != 0:
            raise RuntimeError(f"finalize_agg failed ({rc})")
        out = ctypes.string_at(buf.data, buf.len)
        self._lib.qw_oracle_buf_free(ctypes.byref(buf))
        return json.loads(out)


class GpuSearcher(_BaseSearcher):
    """The product path. Fails loudly when no MI355X is visible
    (QW_ERR_NO_GPU) — there is no CPU fallback."""

    _syms = {"create": "qw_ctx_create", "add_split": "qw_ctx_add_split",
             "leaf_search": "qw_leaf_search", "last_error": "qw_last_error",
             "buf_free": "qw_buf_free"}
    _libname = "libquickwit_amd.so"

    def __init__(self, device: int = 0, config: dict | None = None):
        cfg = dict(config or {})
        cfg.setdefault("device", device)
        # benches/tests run the partial-result cache COLD by default
        # (SURVEY §8a "bypass for benchmarking"); the C-ABI default mirrors
        # the reference's SearcherConfig 64 MB. Pass a capacity to enable.
        cfg.setdefault("partial_request_cache_capacity", 0)
        super().__init__(create_arg=json.dumps(cfg).encode())

    def finalize_agg_json(self, blob: bytes, agg_request: dict) -> dict:
        fn = self._lib.qw_finalize_agg_to_json
        fn.argtypes = [ctypes.c_char_p, ctypes.c_size_t, ctypes.c_char_p,
                       ctypes.POINTER(_Buf)]
        buf = _Buf()
        rc = fn(blob, len(blob), json.dumps(agg_request).encode(), ctypes.byref(buf))
        if rc != 0:
            raise RuntimeError(f"finalize_agg failed ({rc})")
        out = ctypes.string_at(buf.data, buf.len)
        self._lib.qw_buf_free(ctypes.byref(buf))
        return json.loads(out)

    def fetch_docs(self, req: dict) -> dict:
        fn = self._lib.qw_fetch_docs
        fn.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_size_t,
                       ctypes.POINTER(_Buf)]
        req_pb = proto.encode("FetchDocsRequest", req)
        buf = _Buf()
        rc = fn(self._ctx, req_pb, len(req_pb), ctypes.byref(buf))
        if rc != 0:
            raise RuntimeError(f"fetch_docs failed ({rc}): {self._err()}")
        out = ctypes.string_at(buf.data, buf.len)
        self._lib.qw_buf_free(ctypes.byref(buf))
        return proto.decode("FetchDocsResponse", out)

    def leaf_list_terms(self, req: dict) -> dict:
        fn = self._lib.qw_leaf_list_terms
        fn.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_size_t,
                       ctypes.POINTER(_Buf)]
        req_pb = proto.encode("LeafListTermsRequest", req)
        buf = _Buf()
        rc = fn(self._ctx, req_pb, len(req_pb), ctypes.byref(buf))
        if rc != 0:
            raise RuntimeError(f"leaf_list_terms failed ({rc}): {self._err()}")
        out = ctypes.string_at(buf.data, buf.len)
        self._lib.qw_buf_free(ctypes.byref(buf))
        return proto.decode("LeafListTermsResponse", out)

    def kernel_stats(self, name: str):
        fn = self._lib.qw_kernel_stats
        fn.argtypes = [ctypes.c_void_p, ctypes.c_char_p,
                       ctypes.POINTER(ctypes.c_double), ctypes.POINTER(ctypes.c_uint64)]
        ms = ctypes.c_double()
        n = ctypes.c_uint64()
        rc = fn(self._ctx, name.encode(), ctypes.byref(ms), ctypes.byref(n))
        if rc != 0:
            raise KeyError(name)
        return ms.value, n.value

    def kernel_stats_reset(self):
        self._lib.qw_kernel_stats_reset.argtypes = [ctypes.c_void_p]
        self._lib.qw_kernel_stats_reset(self._ctx)

    def memory_stats(self):
        """(used_bytes, budget_bytes, splits_bytes) of the ctx HBM
        accounting (SearchPermitProvider memory-budget analog)."""
        fn = self._lib.qw_ctx_memory_stats
        fn.argtypes = [ctypes.c_void_p] + [ctypes.POINTER(ctypes.c_uint64)] * 3
        u, b, s = ctypes.c_uint64(), ctypes.c_uint64(), ctypes.c_uint64()
        rc = fn(self._ctx, ctypes.byref(u), ctypes.byref(b), ctypes.byref(s))
        if rc != 0:
            raise RuntimeError(f"memory_stats failed: {rc}")
        return u.value, b.value, s.value

    def absence_cache_stats(self):
        """(hits, misses, entries) of the negative term cache
        (leaf.rs:761-827 analog)."""
        fn = self._lib.qw_absence_cache_stats
        fn.argtypes = [ctypes.c_void_p] + [ctypes.POINTER(ctypes.c_uint64)] * 3
        h, m, e = ctypes.c_uint64(), ctypes.c_uint64(), ctypes.c_uint64()
        rc = fn(self._ctx, ctypes.byref(h), ctypes.byref(m), ctypes.byref(e))
        if rc != 0:
            raise RuntimeError(f"absence_cache_stats failed: {rc}")
        return h.value, m.value, e.value

    def device_sync(self):
        self._lib.qw_ctx_device_sync.argtypes = [ctypes.c_void_p]
        rc = self._lib.qw_ctx_device_sync(self._ctx)
        if rc != 0:
            raise RuntimeError(f"device_sync failed: {self._err()}")


# ---------------------------------------------------------------- helpers
def make_leaf_request(query_ast: dict | str, schema: dict, splits: list,
                      max_hits: int = 10, sort_fields: list | None = None,
                      aggregation: dict | None = None, start_offset: int = 0,
                      start_timestamp: int | None = None,
                      end_timestamp: int | None = None) -> dict:
    """Build a LeafSearchRequest dict. `splits` = [(split_id, num_docs), ...]
    or [(split_id, num_docs, footer_start, footer_end), ...]."""
    sreq = {
        "index_id_patterns": ["bench-index"],
        "query_ast": query_ast if isinstance(query_ast, str) else json.dumps(query_ast),
        "max_hits": max_hits,
        "start_offset": start_offset,
    }
    if sort_fields:
        sreq["sort_fields"] = sort_fields
    if aggregation is not None:
        sreq["aggregation_request"] = json.dumps(aggregation)
    if start_timestamp is not None:
        sreq["start_timestamp"] = start_timestamp
    if end_timestamp is not None:
        sreq["end_timestamp"] = end_timestamp
    offsets = []
    for s in splits:
        if isinstance(s, dict):  # full SplitIdAndFooterOffsets (ts metadata
            offsets.append(s)    # drives CanSplitDoBetter pruning)
            continue
        sid, ndocs = s[0], s[1]
        e = {"split_id": sid, "num_docs": ndocs}
        if len(s) > 2:
            e["split_footer_start"], e["split_footer_end"] = s[2], s[3]
        offsets.append(e)
    return {
        "search_request": sreq,
        "leaf_requests": [{"doc_mapper_ord": 0, "index_uri_ord": 0,
                           "split_offsets": offsets}],
        "doc_mappers": [json.dumps(schema)],
        "index_uris": ["ram:///bench-index"],
    }
