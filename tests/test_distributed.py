"""Multi-rank merge path on CPU (gloo, world_size=2) — the exact exchange
bench.py runs over RCCL on the 8-GPU node: each rank searches
its own split, responses are all-gathered, rank 0 merges through
qw_merge_leaf_responses. Oracle searchers stand in for the GPU path here
(same protobuf surfaces); nccl/RCCL only swaps the transport."""
import json
import os

import pytest
import torch.multiprocessing as mp

from quickwit_amd import proto, splitgen

NDOCS = 5_000


def _rank_main(rank, world, q, result):
    import torch.distributed as dist

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29511")
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from quickwit_amd.api import OracleSearcher, make_leaf_request
    from quickwit_amd.merge import merge_leaf_responses

    sid = f"synthetic-3-{rank:04d}"
    data = splitgen.generate_split(rank, NDOCS, seed=3)
    s = OracleSearcher()
    s.add_split(sid, data)
    req = make_leaf_request(q, splitgen.HDFS_SCHEMA, [(sid, NDOCS)], max_hits=15,
                            sort_fields=[{"field_name": "_score", "sort_order": 1}])
    resp_pb = s.leaf_search_raw(proto.encode("LeafSearchRequest", req))

    gathered = [None] * world
    dist.all_gather_object(gathered, resp_pb)
    if rank == 0:
        sreq_pb = proto.encode("SearchRequest", req["search_request"])
        merged = merge_leaf_responses(sreq_pb, gathered)
        result.put(proto.decode("LeafSearchResponse", merged))
    dist.destroy_process_group()


def test_two_rank_gloo_merge_equals_single_call():
    import __graft_entry__
    __graft_entry__.build()
    q = {"type": "bool", "should": [
        {"type": "term", "field": "body", "value": "w%05d" % i} for i in range(3)]}

    ctx = mp.get_context("spawn")
    result = ctx.Queue()
    procs = [ctx.Process(target=_rank_main, args=(r, 2, q, result)) for r in range(2)]
    for p in procs:
        p.start()
    merged = result.get(timeout=180)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    # reference: one oracle over both splits in a single call
    from quickwit_amd.api import OracleSearcher, make_leaf_request
    both = OracleSearcher()
    splits = []
    for r in range(2):
        sid = f"synthetic-3-{r:04d}"
        both.add_split(sid, splitgen.generate_split(r, NDOCS, seed=3))
        splits.append((sid, NDOCS))
    req = make_leaf_request(q, splitgen.HDFS_SCHEMA, splits, max_hits=15,
                            sort_fields=[{"field_name": "_score", "sort_order": 1}])
    expected = both.leaf_search(req)

    assert merged["num_hits"] == expected["num_hits"]
    assert ([(h["split_id"], h["doc_id"]) for h in merged["partial_hits"]] ==
            [(h["split_id"], h["doc_id"]) for h in expected["partial_hits"]])
