"""Decode-kernel variant shoot-out (one gpurun call): times the dominant
kernel of the flagship workloads across library variants x env toggles.

Variants are whole product .so builds (QW_PRODUCT_LIB): the default
TILE_DOCS=8192 and a -DQW_TILE_DOCS=4096 build; env toggles kill the LDS
norms/ktab staging so each lever is measured separately. Prints one line
per (variant, toggles, workload): main-kernel ms/launch + p50 leaf ms.
"""
import argparse
import json
import os
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def run_one(lib, env_toggles, docs, workload):
    env = dict(os.environ, QW_PRODUCT_LIB=lib, **env_toggles)
    code = f"""
import json, sys, time
sys.path.insert(0, {REPO!r})
from quickwit_amd import proto, splitgen
from quickwit_amd.api import GpuSearcher, make_leaf_request
import bench
wl = bench.make_workload({workload!r}, {docs}, 10)
data = bench.cached_split(0, {docs})
s = GpuSearcher(device=0)
s.add_split("synthetic-42-0000", data)
req = make_leaf_request(wl["query"], splitgen.HDFS_SCHEMA,
                        [("synthetic-42-0000", {docs})],
                        max_hits=wl["max_hits"], sort_fields=wl["sort"],
                        aggregation=wl["aggregation"])
pb = proto.encode("LeafSearchRequest", req)
for _ in range(5):
    s.leaf_search_raw(pb)
s.kernel_stats_reset()
ts = []
for _ in range(20):
    t0 = time.perf_counter()
    s.leaf_search_raw(pb)
    ts.append(time.perf_counter() - t0)
ms, n = s.kernel_stats(wl["kernel"])
resp = proto.decode("LeafSearchResponse", s.leaf_search_raw(pb))
print(json.dumps({{"kernel_ms": round(ms / n, 4), "launches": n,
                  "p50_ms": round(sorted(ts)[len(ts)//2] * 1e3, 3),
                  "num_hits": resp.get("num_hits", 0)}}))
"""
    out = subprocess.run([sys.executable, "-c", code], env=env,
                         capture_output=True, text=True, timeout=600)
    if out.returncode != 0:
        return {"error": out.stderr.strip()[-300:]}
    return json.loads(out.stdout.strip().splitlines()[-1])


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--docs", type=int, default=100_000_000)
    args = ap.parse_args()

    variants = [("t8192", os.path.join(REPO, "libquickwit_amd.so")),
                ("t4096", os.path.join(REPO, "libqw_t4096.so"))]
    toggles = [("lds_both", {}),
               ("lds_off", {"QW_NO_LDS_NORMS": "1", "QW_NO_LDS_KTAB": "1"}),
               ("lds_ktab_only", {"QW_NO_LDS_NORMS": "1"})]
    for wl in ("bm25", "range"):
        for vname, lib in variants:
            if not os.path.exists(lib):
                print(f"{wl} {vname}: missing {lib}")
                continue
            for tname, env in toggles:
                t0 = time.time()
                r = run_one(lib, env, args.docs, wl)
                print(f"{wl:6s} {vname} {tname:14s} -> {r} "
                      f"({time.time()-t0:.0f}s)", flush=True)


if __name__ == "__main__":
    main()
