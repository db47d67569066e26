"""Minimal HTTP shim over the C-ABI — enough of Quickwit's REST surface to
replay the reference's rest-api-tests golden scenarios unmodified
(quickwit/rest-api-tests/run_tests.py semantics; SURVEY.md §4 calls these
"the main cross-implementation parity tool"). Serving at scale is OUT OF
SCOPE; this exists so the engine-agnostic scenario suites can
drive the engine end-to-end over HTTP.

Endpoints (subset the suites touch):
  POST   /api/v1/indexes            create index (quickwit index config)
  DELETE /api/v1/indexes/{id}
  POST   /api/v1/{id}/ingest        ndjson body; one split per call
  GET/POST /api/v1/_elastic/{id}/_search   ES body {query, aggs, size, sort}

Dynamic mapping (doc_mapping.mode == "dynamic"): unmapped fields are
inferred from the ingested corpus — str -> text(default, positions) + fast
str column, string arrays -> text + multi-valued str fast, non-negative
int -> u64 fast, int -> i64 fast, floats -> f64 fast, bools -> u64 0/1
with doc-mapper type bool, mixed numeric types -> MIXED column. Explicit
mappings cover text/str/u64/i64/f64/datetime/bool/json/object/concatenate
and tag_fields validation. The engine behind the shim is either the
product (GpuSearcher) or the oracle (OracleSearcher) — the replay asserts
identical golden JSON on both.
"""
import json
import time

from . import splitgen
from .api import make_leaf_request


def _parse_java_date(fmt, s):
    """Java/ES date-format pattern subset (yyyy MM dd HH mm ss S+, quoted
    literals, arbitrary separators) -> epoch milliseconds (float: fractional
    seconds keep sub-ms precision)."""
    import datetime as _dt
    fields = {"yyyy": 0, "MM": 1, "dd": 1, "HH": 0, "mm": 0, "ss": 0}
    frac = 0.0
    fi = si = 0
    while fi < len(fmt):
        matched = False
        for tok in ("yyyy", "MM", "dd", "HH", "mm", "ss"):
            if fmt.startswith(tok, fi):
                fields[tok] = int(s[si:si + len(tok)])
                si += len(tok)
                fi += len(tok)
                matched = True
                break
        if matched:
            continue
        if fmt[fi] == "S":
            n = 0
            while fi + n < len(fmt) and fmt[fi + n] == "S":
                n += 1
            frac = int(s[si:si + n]) / (10 ** n)
            si += n
            fi += n
            continue
        if fmt[fi] == "'":
            j = fmt.index("'", fi + 1)
            lit = fmt[fi + 1:j]
            assert s.startswith(lit, si), (fmt, s, si)
            si += len(lit)
            fi = j + 1
            continue
        assert si < len(s) and s[si] == fmt[fi], (fmt, s, si)
        si += 1
        fi += 1
    d = _dt.datetime(fields["yyyy"], fields["MM"], fields["dd"], fields["HH"],
                     fields["mm"], fields["ss"], tzinfo=_dt.timezone.utc)
    return d.timestamp() * 1000.0 + frac * 1000.0


# -------------------------------------------------- ES query DSL -> QueryAst
# (quickwit-query/src/elastic_query_dsl/ subset the scenario suites use)
def es_query_to_ast(q, schema=None):
    def numeric_field(name):
        for f in (schema or {}).get("fields", []):
            if f["name"] == name and f["type"] in ("u64", "i64", "f64"):
                return True
        return False

    def point_or_term(field, value):
        # term/match over a numeric fast-only column is a point range
        # (the reference's FastFieldRangeQuery translation for dynamic
        # fast fields)
        if numeric_field(field):
            v = value if isinstance(value, (int, float)) else float(value)
            return {"type": "range", "field": field,
                    "lower_bound": {"included": v},
                    "upper_bound": {"included": v}}
        return None

    if q is None:
        return {"type": "match_all"}
    if "match_all" in q:
        return {"type": "match_all"}
    if "match_none" in q:
        return {"type": "match_none"}
    if "term" in q:
        [(field, body)] = q["term"].items()
        value = body["value"] if isinstance(body, dict) else body
        pt = point_or_term(field, value)
        if pt:
            return pt
        if isinstance(body, dict) and body.get("case_insensitive"):
            # tokenized (lowercasing) index: fold the needle too, with the
            # same shared table the tokenizer uses
            from .unicode_tables import lower_cp
            value = "".join(chr(lower_cp(ord(c))) for c in str(value))
        return {"type": "term", "field": field, "value": str(value)}
    if "terms" in q:
        [(field, values)] = [(k, v) for k, v in q["terms"].items()
                             if k != "boost"]
        return {"type": "term_set",
                "terms_per_field": {field: sorted(str(v) for v in values)}}
    if "exists" in q:
        body = q["exists"]
        if not isinstance(body, dict):
            raise ValueError("exists query takes an object with a `field` key")
        field = body["field"]
        sub = [f["name"] for f in (schema or {}).get("fields", [])
               if f["name"] == field or f["name"].startswith(field + ".")]
        if field not in sub and sub:
            # object field: present iff any dotted subfield is present
            return {"type": "bool", "should": [
                {"type": "field_presence", "field": s} for s in sub]}
        return {"type": "field_presence", "field": field}
    if "range" in q:
        [(field, body)] = q["range"].items()
        node = {"type": "range", "field": field}
        fmt = body.get("format") if isinstance(body, dict) else None

        def conv(v):
            if fmt is None:
                return v
            # custom date format (Java-style pattern subset): parse to epoch
            # ms, truncated to the column's fast precision exactly like the
            # reference truncates query bounds (es_compat 0007 pins this:
            # gte == lte == ..001999 matches the ..001 document)
            return int(_parse_java_date(fmt, str(v)))

        if "gte" in body:
            node["lower_bound"] = {"included": conv(body["gte"])}
        elif "gt" in body:
            node["lower_bound"] = {"excluded": conv(body["gt"])}
        if "lte" in body:
            node["upper_bound"] = {"included": conv(body["lte"])}
        elif "lt" in body:
            node["upper_bound"] = {"excluded": conv(body["lt"])}
        return node
    if "match_phrase_prefix" in q:
        [(field, body)] = q["match_phrase_prefix"].items()
        text = str(body["query"] if isinstance(body, dict) else body).strip()
        if " " in text:
            raise ValueError(
                "multi-token match_phrase_prefix not supported")
        # single analyzed token: phrase prefix == term prefix == wildcard
        if "*" in text or "?" in text:
            raise ValueError("phrase prefix with wildcard metachars")
        return {"type": "wildcard", "field": field, "value": text + "*"}
    if "match_bool_prefix" in q:
        # every token is a term except the LAST, which matches as a prefix
        [(field, body)] = q["match_bool_prefix"].items()
        text = str(body["query"] if isinstance(body, dict) else body).strip()
        op = (body.get("operator", "or") if isinstance(body, dict)
              else "or").lower()
        words = text.split()
        if not words:
            return {"type": "match_none"}
        last = words[-1]
        if "*" in last or "?" in last:
            raise ValueError("bool prefix with wildcard metachars")
        subs = [{"type": "full_text", "field": field, "text": w,
                 "params": {"mode": {"type": "bool", "operator": "or"}}}
                for w in words[:-1]]
        subs.append({"type": "wildcard", "field": field,
                     "value": last + "*"})
        key = "must" if op == "and" else "should"
        return {"type": "bool", key: subs}
    if "match_phrase" in q:
        [(field, body)] = q["match_phrase"].items()
        text = body["query"] if isinstance(body, dict) else body
        mode = {"type": "phrase"}
        if isinstance(body, dict) and body.get("slop"):
            mode["slop"] = body["slop"]
        return {"type": "full_text", "field": field, "text": str(text),
                "params": {"mode": mode}}
    if "match" in q:
        [(field, body)] = q["match"].items()
        text = body["query"] if isinstance(body, dict) else body
        pt = point_or_term(field, text)
        if pt:
            return pt
        op = (body.get("operator", "or") if isinstance(body, dict) else "or")
        params = {"mode": {"type": "bool", "operator": op.lower()}}
        if isinstance(body, dict) and "zero_terms_query" in body:
            params["zero_terms_query"] = body["zero_terms_query"]
        return {"type": "full_text", "field": field, "text": str(text),
                "params": params}
    if "prefix" in q:
        [(field, body)] = q["prefix"].items()
        value = body["value"] if isinstance(body, dict) else body
        value = str(value)
        if "*" in value or "?" in value:
            raise ValueError("prefix value with wildcard metachars")
        node = {"type": "wildcard", "field": field, "value": value + "*"}
        if isinstance(body, dict) and body.get("case_insensitive"):
            node["case_insensitive"] = True
        return node
    if "wildcard" in q:
        [(field, body)] = q["wildcard"].items()
        value = body["value"] if isinstance(body, dict) else body
        node = {"type": "wildcard", "field": field, "value": str(value)}
        if isinstance(body, dict) and body.get("case_insensitive"):
            node["case_insensitive"] = True
        return node
    if "bool" in q:
        out = {"type": "bool"}
        for clause in ("must", "must_not", "should", "filter"):
            items = q["bool"].get(clause)
            if items is None:
                continue
            if isinstance(items, dict):
                items = [items]
            out[clause] = [es_query_to_ast(i, schema) for i in items]
        msm = q["bool"].get("minimum_should_match")
        if msm is not None:
            n = len(out.get("should", []))
            if isinstance(msm, str) and msm.endswith("%"):
                pct = int(msm[:-1])
                msm = (n * pct) // 100 if pct >= 0 else n + (n * pct) // 100
            msm = int(msm)
            if msm < 0:
                msm = n + msm  # ES negative form: total minus |msm|
            out["minimum_should_match"] = max(msm, 0)
        return out
    if "query_string" in q:
        body = q["query_string"]
        fields = body.get("fields")
        dfield = body.get("default_field")
        lenient = bool(body.get("lenient", False))
        if fields is not None and not isinstance(fields, list):
            raise ValueError("query_string fields must be an array")
        if fields and dfield:
            raise ValueError(
                "query_string cannot take both default_field and fields")
        dfs = fields if fields else ([dfield] if dfield else None)
        if dfs and schema is not None:
            known = {f["name"] for f in schema.get("fields", [])}
            missing = [f for f in dfs if f not in known]
            if missing and not lenient:
                raise ValueError(
                    f"query_string over unknown fields {missing} "
                    f"(lenient is false)")
            dfs = [f for f in dfs if f in known]
            if not dfs:
                # every named field unknown under lenient: the query must
                # STILL parse (a syntax error is 400 even then); bare
                # tokens then match nothing via a field no split carries
                dfs = ["__lenient_no_field__"]
        return {"type": "user_input", "user_text": body["query"],
                "default_fields": dfs}
    if "multi_match" in q:
        # unknown fields contribute nothing (ES semantics, regardless of
        # lenient); slop and multi-token phrase_prefix are declared out
        body = q["multi_match"]
        fields = body.get("fields")
        if isinstance(fields, str):  # ES accepts a single-string fields
            fields = [fields]
        if not fields:
            raise ValueError("multi_match requires a non-empty fields list")
        mtype = body.get("type", "best_fields")
        if body.get("slop"):
            raise ValueError("multi_match slop > 0 not supported")
        if mtype == "phrase_prefix":
            raise ValueError("multi_match phrase_prefix not supported")
        known = {f["name"] for f in (schema or {}).get("fields", [])}
        use = [f for f in fields if f in known] if schema else fields
        subs = []
        for f in use:
            if mtype == "phrase":
                subs.append({"type": "full_text", "field": f,
                             "text": str(body["query"]),
                             "params": {"mode": {"type": "phrase"}}})
            else:
                op = str(body.get("operator", "or")).lower()
                subs.append({"type": "full_text", "field": f,
                             "text": str(body["query"]),
                             "params": {"mode": {"type": "bool",
                                                 "operator": op}}})
        if not subs:
            return {"type": "match_none"}
        if len(subs) == 1:
            return subs[0]
        return {"type": "bool", "should": subs}
    raise ValueError(f"unsupported es query: {list(q)}")


def flatten_doc(d, prefix=""):
    """Nested objects -> dotted paths (dynamic_mapping expand_dots); arrays
    of objects contribute list-valued dotted leaves (multi-valued text)."""
    out = {}

    def add(key, val):
        if key in out:
            prev = out[key]
            if not isinstance(prev, list):
                prev = [prev]
                out[key] = prev
            if isinstance(val, list):
                prev.extend(val)
            else:
                prev.append(val)
        else:
            out[key] = val

    for k, v in d.items():
        key = prefix + k
        if isinstance(v, dict):
            for k2, v2 in flatten_doc(v, key + ".").items():
                add(k2, v2)
        elif isinstance(v, list) and v and all(isinstance(x, dict) for x in v):
            for elem in v:
                for k2, v2 in flatten_doc(elem, key + ".").items():
                    add(k2, v2)
        else:
            add(key, v)
    return out


def _infer_dynamic_fields(explicit_names, doc_batches):
    """Dynamic-mode field inference over the whole corpus (stable across
    splits). Returns a list of schema field dicts."""
    seen = {}
    for docs in doc_batches:
        for d in docs:
            for k, v in d.items():
                if k in explicit_names or v is None:
                    continue
                kinds = seen.setdefault(k, set())
                if isinstance(v, bool):
                    kinds.add("bool")
                elif isinstance(v, int):
                    if v > 0x7FFFFFFFFFFFFFFF:
                        kinds.add("bigint")  # beyond i64: u64-only territory
                    else:
                        kinds.add("int" if v >= 0 else "negint")
                elif isinstance(v, float):
                    kinds.add("float")
                elif isinstance(v, str):
                    kinds.add("str")
                elif isinstance(v, list) and all(isinstance(x, str) for x in v):
                    kinds.add("strlist")  # text-indexable, no fast column
                else:
                    kinds.add("skip")  # mixed arrays/objects: later round
    fields = []
    for name in sorted(seen):
        kinds = seen[name]
        if "skip" in kinds:
            continue
        if kinds == {"bool"}:
            # boolean dynamic column: u64 0/1 storage, doc-mapper type
            # bool so true/false term literals map to 1/0
            fields.append({"name": name, "type": "bool", "fast": True})
        elif kinds <= {"int", "bigint"}:
            fields.append({"name": name, "type": "u64", "fast": True})
        elif kinds <= {"int", "negint"}:
            fields.append({"name": name, "type": "i64", "fast": True})
        elif kinds <= {"int", "negint", "float"}:
            fields.append({"name": name, "type": "f64", "fast": True})
        elif kinds <= {"int", "negint", "float", "bool", "bigint"}:
            # one dynamic field, several typed columns in the reference
            # (u64 beyond i64 + negatives/floats/bools): our MIXED column
            # (typed values + f64-monotonic sort keys)
            fields.append({"name": name, "type": "mixed", "fast": True})
        elif kinds == {"str"}:
            # quickwit dynamic strings: tokenized text index (with
            # positions, record_field_value_positions) + raw str fast
            # column under the same name (dynamic_mapping tokenizer+fast)
            fields.append({"name": name, "type": "text",
                           "tokenizer": "default", "fast": True,
                           "record": "position"})
        elif "strlist" in kinds and kinds <= {"str", "strlist"}:
            # string arrays: tokenized text index + multi-valued str fast
            # column (distinct sorted values per doc)
            fields.append({"name": name, "type": "text",
                           "tokenizer": "default", "fast": True,
                           "multi": True, "record": "position"})
    return fields


class Index:
    def __init__(self, index_id, config, searcher_factory):
        self.index_id = index_id
        self.config = config
        self.searcher_factory = searcher_factory
        self.batches = []          # raw docs, one list per ingest call
        self.searcher = None
        self.schema = None
        self.splits = []           # (split_id, num_docs)

    def _explicit_fields(self):
        fields = []
        ts_field = None
        dm = self.config.get("doc_mapping", {})
        fast_norm = None
        dyn = dm.get("dynamic_mapping", {})
        if isinstance(dyn.get("fast"), dict):
            fast_norm = dyn["fast"].get("normalizer")
        self.fast_normalizer = fast_norm
        flat_mappings = []

        def walk(fms, prefix=""):
            for fm in fms:
                if fm.get("type") == "object":
                    walk(fm.get("field_mappings", []), prefix + fm["name"] + ".")
                else:
                    fm = dict(fm)
                    fm["name"] = prefix + fm["name"]
                    flat_mappings.append(fm)
        walk(dm.get("field_mappings", []))
        # concatenate fields (doc_mapping type "concatenate"): the listed
        # source fields' leaf values, stringified, indexed under ONE text
        # field with the CONCAT field's tokenizer (scenarii/concat_fields)
        self.concat_specs = []
        self.explicit_top = {fm["name"] for fm in flat_mappings}
        for fm in flat_mappings:
            if fm["type"] == "concatenate":
                self.concat_specs.append({
                    "name": fm["name"],
                    "sources": set(fm.get("concatenate_fields", [])),
                    "include_dynamic":
                        bool(fm.get("include_dynamic_fields", False))})
                cf = {"name": fm["name"], "type": "text",
                      "tokenizer": fm.get("tokenizer", "default")}
                if cf["tokenizer"] != "raw":
                    # splitting analyzers need positions: unquoted values
                    # that analyze to several tokens query as slop-0
                    # phrases (1.5 -> "1 5")
                    cf["record"] = "position"
                fields.append(cf)
        for fm in flat_mappings:
            t = fm["type"]
            name = fm["name"]
            if t == "datetime":
                fields.append({"name": name, "type": "datetime",
                               "fast": bool(fm.get("fast", True))})
                if dm.get("timestamp_field") == name:
                    ts_field = name
            elif t in ("u64", "i64"):
                # numeric mappings are queryable without an explicit fast
                # flag (tag_fields/0002 golden: u64 `tag` with no flags
                # answers term queries); our engine's numeric term/range
                # path is the fast column, so default fast = true
                fields.append({"name": name, "type": t,
                               "fast": bool(fm.get("fast", True))})
            elif t == "text" and fm.get("indexed") is False and fm.get("fast"):
                # fast-only text: raw-string str fast column, no inverted
                # index (term queries become exact point ranges)
                fields.append({"name": name, "type": "str", "fast": True})
            elif t == "text":
                f = {"name": name, "type": "text",
                     "tokenizer": fm.get("tokenizer", "default")}
                if fm.get("record") in ("freq", "position"):
                    f["record"] = fm["record"]
                if fm.get("fast"):
                    f["fast"] = True
                fields.append(f)
        return fields, ts_field

    def rebuild(self):
        fields, ts_field = self._explicit_fields()
        explicit = {f["name"] for f in fields}
        flat_batches = [[flatten_doc(d) for d in b] for b in self.batches]
        if getattr(self, "concat_specs", None):
            for b in flat_batches:
                for d in b:
                    self._add_concat_values(d)
        mode = self.config.get("doc_mapping", {}).get("mode", "dynamic")
        if mode == "dynamic":  # quickwit's default mode
            inferred = _infer_dynamic_fields(explicit, flat_batches)
            if self.fast_normalizer == "lowercase":
                for f in inferred:
                    if f.get("fast"):
                        f["fast_normalizer"] = "lowercase"
            fields = fields + inferred
        self.schema = {"timestamp_field": ts_field, "fields": fields,
                       "default_search_fields":
                           self.config.get("search_settings", {})
                               .get("default_search_fields", [])}
        self.searcher = self.searcher_factory()
        self.splits = []
        self.store_bytes = 0
        batches = flat_batches or [[]]  # empty index: one 0-doc split so
        for i, docs in enumerate(batches):  # aggs return shaped empties
            sid = f"{self.index_id}-{i:04d}"
            w = splitgen.SplitWriter(self.schema, sid)
            w.add_documents(docs)
            data = w.finalize()
            self.store_bytes = getattr(self, "store_bytes", 0) + len(data)
            self.searcher.add_split(sid, data)
            entry = {"split_id": sid, "num_docs": len(docs)}
            if ts_field and docs:
                # split time-range metadata (seconds): drives the
                # CanSplitDoBetter pruning exactly like the reference's
                # SplitIdAndFooterOffsets
                ts = [splitgen.parse_datetime_ms(d[ts_field]) // 1000
                      for d in docs if d.get(ts_field) is not None]
                if ts:
                    entry["timestamp_start"] = min(ts)
                    entry["timestamp_end"] = max(ts)
            self.splits.append(entry)

    def _add_concat_values(self, flat):
        def leaves_under(name):
            vals = []
            for k, v in flat.items():
                if k == name or k.startswith(name + "."):
                    vals.extend(v if isinstance(v, list) else [v])
            return vals

        def stringify(v):
            if isinstance(v, bool):
                return "true" if v else "false"
            if isinstance(v, (int, float)):
                return str(v)
            return v if isinstance(v, str) else None

        for spec in self.concat_specs:
            out = []
            for src in spec["sources"]:
                for v in leaves_under(src):
                    sv = stringify(v)
                    if sv is not None:
                        out.append(sv)
            if spec["include_dynamic"]:
                for k, v in flat.items():
                    top = k.split(".", 1)[0]
                    if top in self.explicit_top:
                        continue
                    for item in (v if isinstance(v, list) else [v]):
                        sv = stringify(item)
                        if sv is not None:
                            out.append(sv)
            if out:
                flat[spec["name"]] = out

    def source_doc(self, split_id, doc_id):
        i = int(split_id.rsplit("-", 1)[1])
        return self.batches[i][doc_id] if self.batches else None


def create_app(searcher_factory):
    from fastapi import FastAPI, Request, Response

    app = FastAPI()
    indexes = {}

    @app.post("/api/v1/indexes")
    @app.post("/api/v1/indexes/")
    async def create_index(request: Request):
        cfg = await request.json()
        iid = cfg["index_id"]
        # tag_fields type validation (doc mapper: only raw-tokenizer text,
        # u64 and i64 fields may be tags — scenarii/tag_fields/0001)
        dm = cfg.get("doc_mapping", {})
        tags = set(dm.get("tag_fields", []))
        if tags:
            by_name = {}

            def walk(fms, prefix=""):
                for fm in fms:
                    if fm.get("type") == "object":
                        walk(fm.get("field_mappings", []),
                             prefix + fm["name"] + ".")
                    else:
                        by_name[prefix + fm["name"]] = fm
            walk(dm.get("field_mappings", []))
            for t in tags:
                fm = by_name.get(t)
                ok = fm is not None and (
                    fm["type"] in ("u64", "i64") or
                    (fm["type"] == "text" and
                     fm.get("tokenizer", "default") == "raw"))
                if not ok:
                    from fastapi.responses import JSONResponse
                    return JSONResponse(status_code=400, content={
                        "message": f"tag field {t} must be raw text, u64 "
                                   f"or i64"})
        idx = Index(iid, cfg, searcher_factory)
        idx.rebuild()
        indexes[iid] = idx
        return {"index_config": {"index_id": iid}}

    @app.delete("/api/v1/indexes/{iid}")
    async def delete_index(iid: str, response: Response):
        if iid not in indexes:
            response.status_code = 404
            return {"message": f"index {iid} not found"}
        del indexes[iid]
        return {"removed": iid}

    @app.delete("/api/v1/_elastic/{targets}")
    @app.delete("/api/v1/{targets}")
    async def es_delete_indices(targets: str, request: Request,
                                response: Response):
        names = [t for t in targets.split(",") if t]
        ignore = request.query_params.get("ignore_unavailable") == "true"
        missing = [n for n in names if n not in indexes]
        if missing and not ignore:
            response.status_code = 404
            return {"message": f"index(es) not found: {missing}"}
        for n in names:
            indexes.pop(n, None)
        return {"acknowledged": True}

    def _stats_payload(matching):
        def one(idxs):
            docs = sum(sp["num_docs"] for ix in idxs for sp in ix.splits)
            size = sum(getattr(ix, "store_bytes", 0) for ix in idxs)
            segs = sum(len(ix.splits) for ix in idxs)
            half = {"docs": {"count": docs, "deleted": 0},
                    "store": {"size_in_bytes": size},
                    "segments": {"count": segs}}
            return {"primaries": half, "total": half}
        return {"_all": one(matching),
                "indices": {ix.index_id: one([ix]) for ix in matching}}

    def _match_indexes(pattern):
        import fnmatch
        return [ix for name, ix in indexes.items()
                if fnmatch.fnmatchcase(name, pattern)]

    @app.get("/api/v1/_stats")
    async def stats_all():
        return _stats_payload(list(indexes.values()))

    @app.get("/api/v1/{pattern}/_stats")
    async def stats_pattern(pattern: str, response: Response):
        matching = _match_indexes(pattern)
        if not matching:
            response.status_code = 404
            return {"message": "no matching index"}
        return _stats_payload(matching)

    @app.post("/api/v1/{iid}/ingest")
    async def ingest(iid: str, request: Request, response: Response):
        if iid not in indexes:
            response.status_code = 404
            return {"message": "index not found"}
        body = (await request.body()).decode()
        docs = [json.loads(line) for line in body.splitlines() if line.strip()]
        idx = indexes[iid]
        # docs missing the configured timestamp field are rejected at ingest
        # (the reference's doc processor requires it)
        ts_field = idx.config.get("doc_mapping", {}).get("timestamp_field")
        if ts_field:
            docs = [d for d in docs if d.get(ts_field) is not None]
        idx.batches.append(docs)
        idx.rebuild()  # commit=force semantics: searchable immediately
        return {"num_docs_for_processing": len(docs)}

    @app.api_route("/api/v1/{iid}/search", methods=["GET", "POST"])
    async def qw_search(iid: str, request: Request, response: Response):
        """Quickwit-native search API subset: ?query=...&start_timestamp=..
        (query strings go through the engine's tantivy-grammar parser)."""
        if iid not in indexes:
            response.status_code = 404
            return {"message": "index not found"}
        idx = indexes[iid]
        params = dict(request.query_params)
        raw = await request.body()
        if raw:
            params.update(json.loads(raw))
        t0 = time.perf_counter()
        ast = {"type": "user_input", "user_text": params.get("query", "*"),
               "default_fields": None}
        req = make_leaf_request(
            ast, idx.schema, idx.splits,
            max_hits=int(params.get("max_hits", 20)),
            start_timestamp=(int(params["start_timestamp"])
                             if "start_timestamp" in params else None),
            end_timestamp=(int(params["end_timestamp"])
                           if "end_timestamp" in params else None))
        resp = idx.searcher.leaf_search(req)
        if resp.get("failed_splits"):
            response.status_code = 400
            return {"message": resp["failed_splits"][0].get("error", "")}
        hits = []
        for h in resp.get("partial_hits", []):
            sid = h.get("split_id", "")
            hits.append(idx.source_doc(sid, h.get("doc_id", 0)))
        return {"num_hits": resp.get("num_hits", 0), "hits": hits,
                "elapsed_time_micros": int((time.perf_counter() - t0) * 1e6),
                "errors": []}

    @app.api_route("/api/v1/{iid}/_count", methods=["GET", "POST"])
    @app.api_route("/api/v1/_elastic/{iid}/_count", methods=["GET", "POST"])
    async def es_count(iid: str, request: Request, response: Response):
        if iid not in indexes:
            response.status_code = 404
            return {"message": "index not found"}
        idx = indexes[iid]
        body = {}
        raw = await request.body()
        if raw:
            body = json.loads(raw)
        if "q" in request.query_params:
            ast = {"type": "user_input",
                   "user_text": request.query_params["q"],
                   "default_fields": None}
        else:
            ast = es_query_to_ast(body.get("query"), idx.schema)
        req = make_leaf_request(ast, idx.schema, idx.splits, max_hits=0)
        resp = idx.searcher.leaf_search(req)
        return {"count": resp.get("num_hits", 0)}

    @app.api_route("/api/v1/{iid}/_search", methods=["GET", "POST"])
    @app.api_route("/api/v1/_elastic/{iid}/_search", methods=["GET", "POST"])
    async def es_search(iid: str, request: Request, response: Response):
        if iid not in indexes:
            response.status_code = 404
            return {"message": "index not found"}
        idx = indexes[iid]
        body = {}
        raw = await request.body()
        if raw:
            body = json.loads(raw)
        t0 = time.perf_counter()
        try:
            qparam = request.query_params.get("q")
            if qparam is not None:  # URL q wins over the body query (0008)
                ast = {"type": "user_input", "user_text": qparam,
                       "default_fields": None} if qparam != "*" else                     {"type": "match_all"}
            else:
                ast = es_query_to_ast(body.get("query"), idx.schema)
        except ValueError as e:
            response.status_code = 400
            return {"message": str(e)}
        extra = request.query_params.get("extra_filters")
        if extra:
            # comma-separated query-string filters ANDed onto the query
            # (es_compat 0023: permissions-style extra filters)
            ast = {"type": "bool", "must": [ast], "filter": [
                {"type": "user_input", "user_text": part,
                 "default_fields": None}
                for part in extra.split(",") if part]}
        aggs = body.get("aggs") or body.get("aggregations")
        size = int(request.query_params.get("size", body.get("size", 10)))
        sorts_in = body.get("sort", [])
        if isinstance(sorts_in, dict):  # ES also accepts a single dict
            sorts_in = [{k: v} for k, v in sorts_in.items()]
        sparam = request.query_params.get("sort")
        if sparam and not sorts_in:
            # ES URL sort syntax: "field:desc,other:asc"
            sorts_in = []
            for part in sparam.split(","):
                f, _, o = part.partition(":")
                sorts_in.append({f: {"order": o or "asc"}})
        sort_fields = []
        for s in sorts_in:
            if isinstance(s, str):
                field, order = s, "asc"
            else:
                [(field, so)] = s.items()
                order = so.get("order", "asc") if isinstance(so, dict) else so
            if field == "_score" and order == "desc":
                sort_fields.append({"field_name": "_score", "sort_order": 1})
            elif field == "_doc":
                # ES _doc = internal doc order: realized as an absent sort
                # column whose doc-id tie-break runs in the sort direction
                # (0008-sort_by golden pins both directions)
                sort_fields.append({"field_name": "_doc",
                                    "sort_order": 1 if order == "desc"
                                    else 0})
            else:
                sort_fields.append({"field_name": field,
                                    "sort_order": 1 if order == "desc" else 0})
        req = make_leaf_request(ast, idx.schema, idx.splits, max_hits=size,
                                sort_fields=sort_fields or None,
                                aggregation=aggs)
        sa = body.get("search_after")
        if sa:
            # datetime sort keys travel as epoch NANOS on the engine's wire
            # (quickwit's internal representation); the ES layer's default
            # input format for datetime search_after literals is epoch
            # millis (0018-search_after golden) unless the sort spec names
            # epoch_nanos_int
            def dt_scale(pos):
                if pos >= len(sorts_in):
                    return 1
                spec = sorts_in[pos]
                if isinstance(spec, str):  # plain "field" sort form
                    spec = {spec: {}}
                if not isinstance(spec, dict):
                    return 1
                [(fname, so)] = spec.items()
                fmt = so.get("format") if isinstance(so, dict) else None
                is_dt = any(f["name"] == fname and f["type"] == "datetime"
                            for f in idx.schema["fields"])
                if not is_dt:
                    return 1
                return 1 if fmt == "epoch_nanos_int" else 1_000_000
            sa = [v * dt_scale(k) if isinstance(v, (int, float))
                  and not isinstance(v, bool) else v
                  for k, v in enumerate(sa)]
            sa = [int(v) * dt_scale(k) if isinstance(v, str)
                  and v.lstrip("-").isdigit() and dt_scale(k) != 1 else v
                  for k, v in enumerate(sa)]
            # ES search_after literal array -> typed SortByValue cursor
            # (SearchAfterSegment conversion happens engine-side)
            def lit_to_sv(x):
                if isinstance(x, bool):
                    return {"boolean": x}
                if isinstance(x, str):
                    # ES coerces string literals for numeric sort keys
                    try:
                        x = int(x)
                    except ValueError:
                        x = float(x)
                if isinstance(x, float):
                    return {"f64": x}
                if isinstance(x, int):
                    return {"i64": x} if x < 0 else {"u64": x}
                raise ValueError(f"unsupported search_after literal: {x!r}")
            cursor = {}
            if len(sa) >= 1:
                cursor["sort_value"] = lit_to_sv(sa[0])
            if len(sa) >= 2:
                cursor["sort_value2"] = lit_to_sv(sa[1])
            req["search_request"]["search_after"] = cursor
        def src_paths(name):
            vals = []
            for v in request.query_params.getlist(name):
                vals.extend(p for p in v.split(",") if p)
            return vals
        src_inc = src_paths("_source_includes")
        src_exc = src_paths("_source_excludes")
        resp = idx.searcher.leaf_search(req)
        if ast.get("type") == "user_input":
            # ES rejects an unparsable query_string with 400; engine-side
            # the parse error is per-split data, so lift it back out when
            # NO split could even parse the query
            failed = resp.get("failed_splits", [])
            if failed and not resp.get("num_successful_splits"):
                err = failed[0].get("error", "")
                if "grammar" in err or "parse" in err or "unknown" in err:
                    response.status_code = 400
                    return {"message": err}
        took_ms = int((time.perf_counter() - t0) * 1e3)
        hits = []
        for h in resp.get("partial_hits", []):
            doc_id = h.get("doc_id", 0)
            sid = h.get("split_id", "")
            src = idx.source_doc(sid, doc_id)
            if src_inc or src_exc:
                src = _filter_source(src, src_inc, src_exc)
            hit = {"_index": iid, "_id": f"{sid}:{doc_id}", "_source": src}
            sv = h.get("sort_value", {})
            if "f64" in sv and sort_fields and \
                    sort_fields[0]["field_name"] == "_score":
                hit["_score"] = sv["f64"]
            if sort_fields:
                def sv_raw(v):
                    for k in ("u64", "i64", "f64", "boolean"):
                        if k in v:
                            return v[k]
                    return None
                srt = [sv_raw(h.get("sort_value", {}))]
                if len(sort_fields) > 1:
                    srt.append(sv_raw(h.get("sort_value2", {})))
                hit["sort"] = srt
            hits.append(hit)
        out = {"took": took_ms, "timed_out": False,
               "hits": {"total": {"value": resp.get("num_hits", 0),
                                  "relation": "eq"},
                        "max_score": None, "hits": hits}}
        if aggs is not None and "intermediate_aggregation_result" in resp:
            out["aggregations"] = idx.searcher.finalize_agg_json(
                resp["intermediate_aggregation_result"], aggs)
        elif aggs is not None:
            out["aggregations"] = {}
        return out

    return app


def _filter_source(src, includes, excludes):
    """ES _source_includes/_source_excludes tree filtering over dotted
    paths (0022-source golden): includes keep matching subtrees, then
    excludes drop theirs."""
    def keep(tree, path):
        if not isinstance(tree, dict) or not path:
            return tree
        head, _, rest = path.partition(".")
        if head not in tree:
            return None
        sub = keep(tree[head], rest) if rest else tree[head]
        return None if sub is None else {head: sub}

    def merge(a, b):
        for k, v in b.items():
            if k in a and isinstance(a[k], dict) and isinstance(v, dict):
                merge(a[k], v)
            else:
                a[k] = v
        return a

    out = src
    if includes:
        out = {}
        for p in includes:
            kept = keep(src, p)
            if kept:
                merge(out, kept)
    if excludes:
        import copy
        out = copy.deepcopy(out)
        for p in excludes:
            parts = p.split(".")
            node = out
            for part in parts[:-1]:
                node = node.get(part) if isinstance(node, dict) else None
                if node is None:
                    break
            if isinstance(node, dict):
                node.pop(parts[-1], None)
    return out
