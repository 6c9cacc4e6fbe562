#!/usr/bin/env python3
"""bench.py — TPC-H Q3 @ SF100 on the MI355X executor (BASELINE.json metric:
rows/sec + GB/s scanned).

One step = one full Q3 pass (customer set build → orders build [+ RCCL
Motions at N>1] → lineitem scan+probe+agg → extract) over AOCS streams
already resident in HBM.  Synthetic TPC-H-shaped tables (no network), hash-
distributed across ranks by the reference's own DISTRIBUTED BY keys; N>1 is
STRONG scaling (fixed SF split across segments, the reference's MPP model).

Single: python bench.py --gpus 1 --steps K --warmup W
Multi:  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
            --master-addr 127.0.0.1 bench.py --gpus N ...
"""
import argparse
import json
import os
import sys
import time

ROOT = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, ROOT)

HBM_PEAK_GBPS = 8000.0   # MI355X HBM3E spec peak (MI355X_MICROARCH.md)


def log(rank, *a):
    if rank == 0:
        print(*a, file=sys.stderr, flush=True)


def _usable_cores():
    """Honest thread count: cgroup cpu quota if set, else affinity."""
    try:
        with open("/sys/fs/cgroup/cpu.max") as f:
            quota, period = f.read().split()
            if quota != "max":
                return max(1, int(int(quota) / int(period)))
    except OSError:
        pass
    return len(os.sched_getaffinity(0))


def cpu_baseline_leg(sample_sf=8.0, full=False):
    """Oracle (CPU restatement, 'port', OpenMP) timed on this host: the
    reported CPU baseline (BASELINE.md — the reference publishes no
    numbers).  Bounded sample by default; --full-cpu-baseline times the
    COMPLETE workload SF instead (one pass, no extrapolation — VERDICT r01
    weak #4)."""
    cores = int(os.environ.get("GX_CPU_THREADS", _usable_cores()))
    from oracle import pyapi as orc   # checker/baseline use only
    c = orc.gen_customer(sample_sf)
    o = orc.gen_orders(sample_sf)
    li = orc.gen_lineitem(sample_sf)
    if full:
        orc.set_threads(cores)
        t0 = time.perf_counter()
        orc.q3(c, o, li)
        dt = time.perf_counter() - t0
        rows = len(li["l_orderkey"])
        return {"value": rows / dt, "unit": "rows/s", "cores": cores,
                "kind": "port",
                "sample": f"tpch_q3_sf{sample_sf:g} oracle pipeline FULL x1 "
                          f"({rows} lineitem rows, {cores} OpenMP threads)"}
    # calibrate: cgroup quotas can make "nproc" threads slower than fewer —
    # pick the faster of {1, cores/2, cores} on one pass and report that count
    best, cores_eff = None, 1
    for t_try in sorted({1, max(1, cores // 2), cores}):
        orc.set_threads(t_try)
        t0 = time.perf_counter()
        orc.q3(c, o, li)
        dt = time.perf_counter() - t0
        if best is None or dt < best:
            best, cores_eff = dt, t_try
    orc.set_threads(cores_eff)
    cores = cores_eff
    reps, t = 0, 0.0
    t_end = time.time() + 12.0
    while time.time() < t_end or reps < 2:
        t0 = time.perf_counter()
        orc.q3(c, o, li)
        t += time.perf_counter() - t0
        reps += 1
        if reps >= 8:
            break
    rows = len(li["l_orderkey"]) * reps
    return {"value": rows / t, "unit": "rows/s", "cores": cores, "kind": "port",
            "sample": f"tpch_q3_sf{sample_sf:g} oracle pipeline x{reps} "
                      f"({len(li['l_orderkey'])} lineitem rows/pass, "
                      f"{cores} OpenMP threads)"}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--sf", type=float, default=100.0)
    ap.add_argument("--no-cpu-baseline", action="store_true")
    ap.add_argument("--full-cpu-baseline", action="store_true",
                    help="time the oracle on the FULL workload SF (one pass, "
                         "no extrapolation) instead of the bounded sample")
    ap.add_argument("--no-traffic", action="store_true",
                    help="skip the rocprofv3 --pmc FETCH_SIZE side-run that "
                         "fills roofline.traffic (used by the side-run "
                         "itself to avoid recursion)")
    ap.add_argument("--rle-keys", action="store_true",
                    help="store l_orderkey RLE-compressed (rle_type); the fused "
                         "probe kernel then scans runs, not rows (extra mode — "
                         "the judged default stays compresstype=none)")
    ap.add_argument("--force-motion", action="store_true",
                    help="route the run through the FULL RCCL exchange branch "
                         "even at 1 rank (self send/recv) — the multi-GPU code "
                         "path, testable on one GPU (extra mode)")
    args = ap.parse_args()
    if args.force_motion:
        os.environ["GX_FORCE_MOTION"] = "1"

    # the contract is ONE JSON line on stdout from rank 0; libraries (RCCL
    # prints a version banner at communicator init) must not pollute it —
    # route C-level stdout to stderr for the whole run and restore the real
    # stdout only for the final JSON line.
    real_stdout = os.dup(1)
    os.dup2(2, 1)

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    # --gpus N must match the actual process count: a 1-process run may not
    # stamp n_gpus>1 on the record
    assert max(world, 1) == args.gpus, \
        f"WORLD_SIZE={world} but --gpus={args.gpus}: launch one rank per GPU " \
        f"(torch.distributed.run --nproc-per-node {args.gpus})"
    n = args.gpus

    import numpy as np
    import torch
    import cloudberry_amd as gx

    dist = None
    if n > 1:
        import torch.distributed as tdist
        tdist.init_process_group(backend="gloo")  # bootstrap/barriers only;
        dist = tdist                              # data path is our own RCCL comm

    ndev = max(torch.cuda.device_count(), 1)
    device = local_rank % ndev       # oversubscription only for dev testing
    ctx = gx.Context(device=device, seg=rank, nsegs=n)
    torch.cuda.set_device(device)

    if n > 1:
        # broadcast RCCL unique id over gloo, then build the communicator
        if rank == 0:
            uid = ctx.comm_unique_id()
            t = torch.tensor(list(uid), dtype=torch.uint8)
        else:
            t = torch.zeros(128, dtype=torch.uint8)
        dist.broadcast(t, src=0)
        ctx.comm_init(bytes(t.tolist()))
    elif args.force_motion:
        ctx.comm_init(ctx.comm_unique_id())   # 1-rank communicator

    # ---- setup (untimed): generate + AOCS-encode this segment's shard ----
    t0 = time.time()
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, args.sf)
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, args.sf)
    li_kind = gx.TPCH_LINEITEM_RLEKEY if args.rle_keys else gx.TPCH_LINEITEM
    li = ctx.tpch_gen(li_kind, args.sf)
    q = ctx.q3(cust, ordr, li)
    log(rank, f"setup: sf={args.sf} rank {rank}/{n}: "
              f"{cust.nrows} cust, {ordr.nrows} ord, {li.nrows} li rows "
              f"({time.time()-t0:.1f}s)")

    li_rows_local = li.nrows
    bytes_local = cust.logical_bytes + ordr.logical_bytes + li.logical_bytes
    if n > 1:
        t = torch.tensor([li_rows_local, int(bytes_local)], dtype=torch.int64)
        dist.all_reduce(t)
        li_rows_total, bytes_total = int(t[0]), float(t[1])
    else:
        li_rows_total, bytes_total = li_rows_local, bytes_local

    # ---- warmup ----
    for _ in range(args.warmup):
        q.run()

    # ---- timed region ----
    if dist: dist.barrier()
    torch.cuda.synchronize()
    t_start = time.perf_counter()
    probe_ms = 0.0
    for _ in range(args.steps):
        q.run()
        probe_ms += q.stats()["ms_probe_agg"]
    torch.cuda.synchronize()
    if dist: dist.barrier()
    elapsed = time.perf_counter() - t_start

    # MAX over ranks
    if n > 1:
        t = torch.tensor([elapsed])
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t[0])

    st = q.stats()
    ms_step = elapsed * 1000.0 / args.steps
    rows_per_sec = li_rows_total / (elapsed / args.steps)
    gbps = bytes_total / 1e9 / (elapsed / args.steps)

    # roofline: dominant kernel = fused lineitem scan+probe+agg.
    # algorithmic bytes/launch = lineitem 4 cols (28 B/row) of
    # THIS rank; duration from HIP events on the executor stream.
    probe_ms_avg = probe_ms / args.steps
    li_bytes_local = li_rows_local * 28.0
    achieved = li_bytes_local / 1e9 / (probe_ms_avg / 1000.0)
    # traffic: DRAM-side read bytes of the dominant kernel from a separate
    # rocprofv3 --pmc FETCH_SIZE pass (counters only), with the documented
    # gfx950 x2 correction — tools/pmc_traffic.py.  Fail-soft: null.
    traffic = None
    if rank == 0 and n == 1 and not args.no_traffic:
        try:
            from tools.pmc_traffic import probe_kernel_fetch_bytes
            log(rank, "PMC side-run (rocprofv3 --pmc FETCH_SIZE)...")
            r = probe_kernel_fetch_bytes(args.sf)
            if r:
                traffic = round(r[0], 0)
                log(rank, f"  FETCH_SIZE {r[1]/1048576.0:.2f} GB/launch raw "
                          f"x2 -> {traffic/1e9:.2f} GB over {r[2]} dispatches")
        except Exception as e:
            log(rank, f"  traffic probe failed: {e}")
    roofline = {"bound": "hbm", "achieved": round(achieved, 1),
                "peak": HBM_PEAK_GBPS, "unit": "GB/s",
                "frac": round(achieved / HBM_PEAK_GBPS, 4),
                "traffic": traffic}

    # config-2 evidence: the pure scan+filter kernel's GB/s on the widest
    # filter column (untimed extra; printed into config below)
    scan_gbps = None
    if rank == 0 and n == 1:
        cnt, sf_ms = li.scan_filter(3, ">", gx.CUTOFF_19950315)
        if sf_ms > 0:
            scan_gbps = round(li_rows_local * 4.0 / 1e9 / (sf_ms / 1000.0), 1)

    cpu = None
    if rank == 0 and n == 1 and not args.no_cpu_baseline:
        if args.full_cpu_baseline:
            log(rank, f"timing CPU baseline (oracle, FULL sf={args.sf:g})...")
            cpu = cpu_baseline_leg(args.sf, full=True)
        else:
            log(rank, "timing CPU baseline (oracle, bounded sample)...")
            cpu = cpu_baseline_leg()

    if rank == 0:
        out = {
            "metric": "tpch_q3_rows_per_sec",
            "value": round(rows_per_sec, 1),
            "unit": "rows/s",
            "n_gpus": n,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_step, 3),
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "f64",
            "data": "synthetic",
            "config": {"workload": f"tpch_q3_sf{args.sf:g}", "sf": args.sf,
                       "lineitem_rows": li_rows_total,
                       "scan_bytes": bytes_total,
                       "gbps_scanned": round(gbps, 1),
                       "parallelism": f"mpp{n}",
                       "lineitem_key_format": "rle_type" if args.rle_keys else "none",
                       "scan_filter_kernel_gbps": scan_gbps,
                       "groups": st["groups"],
                       "stage_ms": {k: round(st[k], 3) for k in
                                    ("ms_cust_build", "ms_orders_build",
                                     "ms_motion", "ms_motion_counts",
                                     "ms_motion_payload",
                                     "ms_probe_agg", "ms_extract")}},
            "roofline": roofline,
            "cpu_baseline": cpu,
        }
        sys.stdout.flush()
        os.dup2(real_stdout, 1)
        print(json.dumps(out), flush=True)
        os.dup2(2, 1)          # teardown prints (RCCL banner) go to stderr

    q.free(); li.free(); ordr.free(); cust.free(); ctx.close()
    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
