#!/usr/bin/env python3
"""Extract golden parity vectors from the reference checkout into JSON fixtures.

Runs ONLY in the build container, where /root/reference (quickwit-oss/quickwit,
workspace 0.9.0) is mounted read-only. The GPU box never sees /root/reference;
tests consume the committed JSON files in this directory instead.

Sources:
  - BM25 exact scores: quickwit/quickwit-search/src/tests.rs:600-691
    (test_sort_bm25) — transcribed below with the corpus and expected
    (score_f32, doc_id) lists, since Rust cannot be parsed/compiled here.
  - Aggregation buckets: quickwit/rest-api-tests/scenarii/aggregations/
    {_setup.quickwit.yaml, 0001-aggregations.yaml} — parsed from YAML.
"""
import json
import os
import sys

REF = "/root/reference/quickwit"
OUT = os.path.dirname(os.path.abspath(__file__))


def bm25_golden():
    # Transcription of test_sort_bm25 (tests.rs:600-691).
    # Doc mapping (tests.rs:602-617): title text record=freq fieldnorms,
    # nofreq text record=basic fieldnorms. Default search fields include both.
    docs = [
        {"title": "one pad", "nofreq": "two pad"},  # doc 0
        {"title": "one", "nofreq": "two"},          # doc 1
        {"title": "one one", "nofreq": "two two"},  # doc 2
    ]
    cases = [
        # (query_ast, expected [(score_f32, doc_id)] sorted desc)  tests.rs:670-688
        {
            "name": "term_with_freq",
            "query_ast": {"type": "term", "field": "title", "value": "one"},
            "expected": [[0.1738279, 2], [0.15965714, 1], [0.12343242, 0]],
        },
        {
            "name": "term_record_basic",
            "query_ast": {"type": "term", "field": "nofreq", "value": "two"},
            "expected": [[0.15965714, 1], [0.12343242, 2], [0.12343242, 0]],
        },
        {
            "name": "two_term_disjunction",
            "query_ast": {
                "type": "bool",
                "should": [
                    {"type": "term", "field": "title", "value": "one"},
                    {"type": "term", "field": "nofreq", "value": "two"},
                ],
            },
            "expected": [[0.31931427, 1], [0.2972603, 2], [0.24686484, 0]],
        },
    ]
    return {
        "source": "quickwit/quickwit-search/src/tests.rs:600-691 (test_sort_bm25)",
        "sort": ["_score desc"],
        "schema": [
            {"name": "title", "type": "text", "record": "freq", "fieldnorms": True},
            {"name": "nofreq", "type": "text", "record": "basic", "fieldnorms": True},
        ],
        "docs": docs,
        "cases": cases,
    }


def agg_golden():
    import yaml

    setup_path = os.path.join(REF, "rest-api-tests/scenarii/aggregations/_setup.quickwit.yaml")
    tests_path = os.path.join(REF, "rest-api-tests/scenarii/aggregations/0001-aggregations.yaml")
    setup_steps = list(yaml.safe_load_all(open(setup_path)))
    # The two ingest steps carry the corpus, one per split (commit: force).
    splits = []
    for step in setup_steps:
        if step and step.get("endpoint") == "aggregations/ingest":
            splits.append(step["ndjson"])
    assert len(splits) == 2 and len(splits[0]) == 5 and len(splits[1]) == 5

    wanted = {
        # scenario index (0-based across the YAML docs) -> fixture name; only
        # cases within the round-1 subset (single-valued fields, no
        # percentiles/cardinality/split_size error bounds).
        "date_histogram_basic": 0,
        "date_histogram_extended_bounds": 1,
        "date_histogram_stats_subagg": 2,
        "date_histogram_stats_subagg_exists_filter": 3,
        "terms_full": 5,
        "histogram_interval50": 11,
    }
    steps = list(yaml.safe_load_all(open(tests_path)))
    cases = {}
    for name, idx in wanted.items():
        step = steps[idx]
        cases[name] = {
            "request": step["json"],
            "expected": step["expected"]["aggregations"],
        }
    return {
        "source": "quickwit/rest-api-tests/scenarii/aggregations/{_setup.quickwit.yaml,0001-aggregations.yaml}",
        "schema": [
            {"name": "date", "type": "datetime", "fast": True, "nullable": True},
            {"name": "response", "type": "u64", "fast": True, "nullable": True},
            {"name": "name", "type": "str", "fast": True, "nullable": True},
            {"name": "host", "type": "str", "fast": True, "nullable": True},
            {"name": "id", "type": "u64", "fast": True, "nullable": True},
            {"name": "tags", "type": "text", "tokenizer": "default",
             "fast": True, "multi": True},
        ],
        "splits": splits,
        "cases": cases,
    }


def rest_scenarios():
    """Full step lists of selected rest-api-tests suites, replayed by
    tests/test_rest_scenarios.py against the HTTP shim (quickwit_amd/rest.py)
    with run_tests.py's checking semantics (ported in tests/rest_replay.py)."""
    import yaml

    base = os.path.join(REF, "rest-api-tests/scenarii")
    suites = {
        "aggregations": ["_setup.quickwit.yaml", "0001-aggregations.yaml",
                         "_teardown.quickwit.yaml"],
        "sort_orders": ["_setup.quickwit.yaml", "0001-sort-elasticapi.yaml",
                        "_teardown.quickwit.yaml"],
        "search_after": ["_setup.quickwit.yaml",
                         "0001-search_after_edge_case.yaml",
                         "_teardown.quickwit.yaml"],
        "qw_search_api": ["_setup.quickwit.yaml", "0001_ts_range.yaml",
                          "0002_negative_search.yaml",
                          "0003_exists_search.yaml", "0004_exact_string.yaml",
                          "0005_fast_field_search.yaml",
                          "_teardown.quickwit.yaml"],
        "default_search_fields": ["_setup.quickwit.yaml",
                                  "0001_default_fields.yaml",
                                  "_teardown.quickwit.yaml"],
        "multi_splits": ["_setup.quickwit.yaml",
                         "0001-request-optimizations.yaml",
                         "_teardown.quickwit.yaml"],
        "tag_fields": ["_setup.quickwit.yaml", "0001_allowed_types.yaml",
                       "0002_negative_tags.yaml", "_teardown.quickwit.yaml"],
        "concat_fields": ["_setup.quickwit.yaml",
                          "0001_concat_field.yaml",
                          "_teardown.quickwit.yaml"],
        # the in-scope slice of the big ES-compat suite (gharchive corpus)
        "es_compatibility": ["_setup.quickwit.yaml", "0001-noquery.yaml",
                             "0003-match.yaml", "0006-term_query.yaml",
                             "0007-range_queries.yaml", "0009-bool_query.yaml",
                             "0011-exists-query.yaml", "0015-terms-query.yaml",
                             "0019-count.yaml", "0029-wildcard.yaml",
                             "_teardown.quickwit.yaml"],
        "es_compat_extra": ["_setup.quickwit.yaml", "0008-sort_by.yaml",
                            "0013-phrase-query.yaml", "0018-search_after.yaml",
                            "0020-stats.yaml",
                            "0028-fast_only_field_query.yaml",
                            "0030-prefix.yaml", "0002-query_string.yaml",
                            "0004-term_aggregations.yaml",
                            "0005-query_string_query.yaml",
                            "0010-match_phrase_prefix_query.yaml",
                            "0014-multi-match-query.yaml",
                            "0016-misc-query.yaml",
                            "0017-match-bool-prefix-query.yaml",
                            "0022-source.yaml",
                            "0023-extra_filters.yaml",
                            "0024-delete_indices.yaml",
                            "_teardown.quickwit.yaml"],
    }
    keep = {"method", "endpoint", "params", "json", "ndjson", "expected",
            "status_code", "ndjson_file"}
    out = {}
    # suites whose files live in another suite's directory
    dir_of = {"es_compat_extra": "es_compatibility"}
    for suite, files in suites.items():
        sdir = dir_of.get(suite, suite)
        # suite context defaults (run_tests.py Visitor context stacking)
        ctx = {}
        ctx_path = os.path.join(base, sdir, "_ctx.yaml")
        if os.path.exists(ctx_path):
            ctx = yaml.safe_load(open(ctx_path)) or {}
        ctx_endpoint = ctx.get("endpoint")
        api_root = ctx.get("api_root", "")
        prefix = "_elastic/" if api_root.rstrip("/").endswith("_elastic") else ""
        steps = []
        for fn in files:
            is_scenario = not fn.startswith("_")
            for step in yaml.safe_load_all(open(os.path.join(base, sdir, fn))):
                if not isinstance(step, dict):
                    continue
                if is_scenario:  # numbered scenarios inherit the suite ctx
                    if "method" not in step:
                        step["method"] = ctx.get("method", "GET")
                    if "endpoint" not in step and ctx_endpoint:
                        step["endpoint"] = ctx_endpoint
                    if "endpoint" in step and prefix and \
                            not step["endpoint"].startswith("_elastic"):
                        step["endpoint"] = prefix + step["endpoint"]
                if "method" not in step:
                    continue
                engines = step.get("engines")
                if engines and "quickwit" not in engines:
                    continue
                if step.get("endpoint") == "_bulk" and "body_from_file" in step:
                    # ES bulk gz corpus -> committed ndjson fixture + a plain
                    # ingest step per target index
                    import gzip
                    raw = gzip.open(os.path.join(base, sdir,
                                                 step["body_from_file"])).read()
                    lines = [l for l in raw.decode().splitlines() if l.strip()]
                    by_index = {}
                    for k in range(0, len(lines), 2):
                        action = json.loads(lines[k])
                        target = action.get("index", {}).get("_index", "")
                        by_index.setdefault(target, []).append(lines[k + 1])
                    for target, docs in by_index.items():
                        fx = f"{suite}-{target}.ndjson.gz"
                        with gzip.open(os.path.join(OUT, fx), "wb") as g:
                            g.write(("\n".join(docs) + "\n").encode())
                        steps.append({"method": "POST",
                                      "endpoint": f"{target}/ingest",
                                      "params": {"commit": "force"},
                                      "ndjson_file": fx})
                    continue
                s = {k: v for k, v in step.items() if k in keep}
                if "shuffle_ndjson" in step:
                    # split distribution must not affect results: resolve the
                    # randomized distribution ONCE at extraction (seed 42,
                    # same bucketing as run_tests.py distribute_items)
                    import random
                    rng = random.Random(42)
                    nb = rng.randint(step.get("min_splits", 1),
                                     step.get("max_splits", 5))
                    buckets = [[] for _ in range(nb)]
                    for item in step["shuffle_ndjson"]:
                        buckets[rng.randint(0, nb - 1)].append(item)
                    for b in buckets:
                        if not b:
                            continue
                        sb = dict(s)
                        sb["ndjson"] = b
                        if isinstance(sb["method"], list):
                            sb["method"] = sb["method"][0]
                        steps.append(sb)
                    continue
                if isinstance(s["method"], list):
                    s["method"] = s["method"][0]
                if "status_code" in step and step["status_code"] is None:
                    s["status_code"] = None
                steps.append(s)
        out[suite] = steps
    return {
        "source": "quickwit/rest-api-tests/scenarii/ (run_tests.py driver "
                  "semantics ported to tests/rest_replay.py)",
        "suites": out,
    }


def main():
    if not os.path.isdir(REF):
        sys.exit("reference checkout not present; fixtures must already be committed")
    with open(os.path.join(OUT, "bm25_sort.json"), "w") as f:
        json.dump(bm25_golden(), f, indent=1, sort_keys=True)
        f.write("\n")
    with open(os.path.join(OUT, "aggregations.json"), "w") as f:
        json.dump(agg_golden(), f, indent=1, sort_keys=True)
        f.write("\n")
    with open(os.path.join(OUT, "rest_scenarios.json"), "w") as f:
        json.dump(rest_scenarios(), f, indent=1, sort_keys=True)
        f.write("\n")
    print("wrote bm25_sort.json, aggregations.json, rest_scenarios.json")


if __name__ == "__main__":
    main()
