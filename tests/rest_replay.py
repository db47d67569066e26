"""Replay driver for the reference's rest-api-tests scenarios against the
HTTP shim (quickwit_amd/rest.py). Ports run_tests.py's step execution and
response checking semantics (quickwit/rest-api-tests/run_tests.py:109-224):
subset-matching of expected JSON, `$expect` eval hooks, exact list lengths.
"""
import json


def check_result(result, expected, path=""):
    if isinstance(expected, dict) and "$expect" in expected:
        exprs = expected["$expect"]
        if isinstance(exprs, str):
            exprs = [exprs]
        for expr in exprs:
            assert eval(expr, None, {"val": result}), \
                f"$expect failed at {path}: {expr} (val={result!r})"
        return
    assert type(result) == type(expected) or (
        isinstance(result, (int, float)) and isinstance(expected, (int, float))
    ), f"type mismatch at {path}: {type(result)} vs {type(expected)}"
    if isinstance(expected, dict):
        for k, v in expected.items():
            assert k in result, f"missing key `{k}` at {path} (got {list(result)})"
            check_result(result[k], v, f"{path}.{k}")
    elif isinstance(expected, list):
        assert len(result) == len(expected), (
            f"wrong length at {path}: expected {len(expected)} got "
            f"{len(result)}: {json.dumps(result)[:800]}")
        for i, (left, right) in enumerate(zip(result, expected)):
            check_result(left, right, f"{path}[{i}]")
    else:
        assert result == expected, f"expected {expected!r} at {path}, got {result!r}"


def agg_kinds(aggs):
    kinds = set()

    def walk(a):
        for body in a.values():
            for k, v in body.items():
                if k == "aggs":
                    walk(v)
                else:
                    kinds.add((k, v.get("field") if isinstance(v, dict) else None))
    walk(aggs or {})
    return kinds


def run_step(client, step):
    method = step["method"]
    url = "/api/v1/" + step["endpoint"].lstrip("/")
    kwargs = {}
    if "params" in step:
        kwargs["params"] = step["params"]
    if "ndjson_file" in step:
        import gzip
        import os
        path = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                            "golden", step["ndjson_file"])
        kwargs["content"] = gzip.open(path).read()
    elif "ndjson" in step:
        kwargs["content"] = "\n".join(
            json.dumps(d) for d in step["ndjson"]) + "\n"
    elif "json" in step:
        kwargs["json"] = step["json"]
    r = client.request(method, url, **kwargs)
    expected_status = step.get("status_code", 200)
    if expected_status is not None:
        assert r.status_code == expected_status, \
            f"{method} {url}: status {r.status_code} != {expected_status}: {r.text[:500]}"
    if step.get("expected") is not None:
        check_result(r.json(), step["expected"], "")
    return r


def replay_suite(client, steps, skip_step=None):
    """Returns (ran, skipped_reasons)."""
    ran, skipped = 0, []
    for i, step in enumerate(steps):
        reason = skip_step(i, step) if skip_step else None
        if reason:
            skipped.append((i, reason))
            continue
        run_step(client, step)
        ran += 1
    return ran, skipped
