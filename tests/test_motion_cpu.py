"""World-size-2 gloo CPU tests of the multi-segment execution semantics:
two ranks each run their hash-distributed shard (oracle as the per-segment
executor), exchange filtered orders (Motion 1 by o_custkey) and qualifying
orders (Motion 2 by o_orderkey) through torch.distributed, and the unioned
result must equal the global single-segment Q3.  This mirrors the exact
orchestration gx_q3_run performs with RCCL on GPUs."""
import os
import sys

import numpy as np
import torch
import torch.multiprocessing as mp

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SF = 0.02


def _exchange(dist, arrays, dest, nsegs, rank):
    """all-to-all of numpy rows by destination (counts exchange + payload),
    the same two-phase shape as the RCCL path."""
    send_chunks = []
    counts = torch.zeros(nsegs, dtype=torch.int64)
    order = np.argsort(dest, kind="stable")
    for d in range(nsegs):
        m = dest == d
        counts[d] = int(m.sum())
        send_chunks.append({k: torch.from_numpy(np.ascontiguousarray(v[m]))
                            for k, v in arrays.items()})
    # counts all-gather (mirrors ncclAllGather of per-dest counts)
    all_counts = [torch.zeros(nsegs, dtype=torch.int64) for _ in range(nsegs)]
    dist.all_gather(all_counts, counts)
    out = {}
    for k in arrays:
        recv = []
        for src in range(nsegs):
            n = int(all_counts[src][rank])
            buf = torch.zeros(n, dtype=send_chunks[0][k].dtype)
            if src == rank:
                buf = send_chunks[rank][k]
                recv.append(buf)
                continue
            # pairwise send/recv (grouped send/recv analog)
            if rank < src:
                dist.send(send_chunks[src][k], dst=src)
                dist.recv(buf, src=src)
            else:
                dist.recv(buf, src=src)
                dist.send(send_chunks[src][k], dst=src)
            recv.append(buf)
        out[k] = np.concatenate([b.numpy() for b in recv]) if recv else np.array([])
    return out


def _worker(rank, world, result_q):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(29500 + 17 * world)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    sys.path.insert(0, ROOT)
    from oracle import pyapi as orc

    cutoff = orc.CUTOFF_19950315
    c = orc.gen_customer(SF, seg=rank, nsegs=world)
    o = orc.gen_orders(SF, seg=rank, nsegs=world)
    li = orc.gen_lineitem(SF, seg=rank, nsegs=world)

    # Motion 1: filtered orders redistributed by o_custkey
    m = o["o_orderdate"] < cutoff
    dest = orc.route(o["o_custkey"][m], world)
    o1 = _exchange(dist, {k: v[m] for k, v in o.items()}, dest, world, rank)

    # local semijoin against this segment's BUILDING customers
    segok = c["c_custkey"][c["c_mktsegment"] == 0]
    qual = np.isin(o1["o_custkey"], segok)
    # Motion 2: qualifying orders redistributed by o_orderkey
    dest2 = orc.route(o1["o_orderkey"][qual], world)
    o2 = _exchange(dist, {k: v[qual] for k, v in o1.items()}, dest2, world, rank)

    # local probe+agg (customer filter already applied upstream)
    c2 = {"c_custkey": o2["o_custkey"],
          "c_mktsegment": np.zeros(len(o2["o_custkey"]), np.uint8)}
    res = orc.q3(c2, o2, li)
    result_q.put((rank, {k: v for k, v in res.items()}))
    dist.destroy_process_group()


def _run_motion_pipeline(nsegs):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, nsegs, q)) for r in range(nsegs)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(nsegs):
        rank, res = q.get(timeout=300)
        results[rank] = res
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    from oracle import pyapi as orc
    glob = orc.q3(orc.gen_customer(SF), orc.gen_orders(SF), orc.gen_lineitem(SF))
    keys = np.concatenate([results[r]["l_orderkey"] for r in range(nsegs)])
    rev = np.concatenate([results[r]["revenue"] for r in range(nsegs)])
    cnt = np.concatenate([results[r]["nitems"] for r in range(nsegs)])
    order = np.argsort(keys)
    assert len(keys) == len(glob["l_orderkey"])
    assert (keys[order] == glob["l_orderkey"]).all()
    assert (cnt[order] == glob["nitems"]).all()
    np.testing.assert_allclose(rev[order], glob["revenue"], rtol=1e-9)
    # and each rank only produced groups that route to it
    for r in range(nsegs):
        assert (orc.route(results[r]["l_orderkey"], nsegs) == r).all()


def test_two_rank_gloo_motion_equals_global():
    _run_motion_pipeline(2)


def test_four_rank_gloo_motion_equals_global():
    """the exact count layout the 8-rank scale bench uses,
    replayed at ws=4 on CPU (pairwise exchanges, per-rank shard gen)."""
    _run_motion_pipeline(4)


def test_eight_rank_gloo_motion_equals_global():
    """The exact 8-rank topology of the round-end scale bench, replayed on
    CPU: 8 hash-distributed shards, two Motions, unioned result equals the
    global single-segment Q3."""
    _run_motion_pipeline(8)
