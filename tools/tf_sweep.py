#!/usr/bin/env python3
"""Sweep GX_TABLE_FACTOR_PCT (join/agg table slots = qual * tf / 100,
pow2-rounded): the interpolation slot layout trades collision-walk length
(small tables) against Infinity-Cache residency (big tables).  r1 shipped
tf=200 untuned."""
import json
import os
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)
import cloudberry_amd as gx  # noqa: E402

SF = float(os.environ.get("AB_SF", "100"))
REPS = int(os.environ.get("AB_REPS", "6"))


def main():
    ctx = gx.Context(device=0, seg=0, nsegs=1)
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, SF)
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, SF)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM, SF)
    results = {}
    base = None
    for tf in tuple(int(x) for x in os.environ.get("TF_LIST", "200,100,125,150,300,400").split(",")):
        os.environ["GX_TABLE_FACTOR_PCT"] = str(tf)
        q = ctx.q3(cust, ordr, li)     # fresh q: sizing rereads the env
        best = None
        for _ in range(REPS):
            q.run()
            st = q.stats()
            if best is None or st["ms_probe_agg"] < best["ms_probe_agg"]:
                best = st
        if base is None:
            base = best
        ok = (best["probe_hits"] == base["probe_hits"] and
              best["groups"] == base["groups"])
        print(f"tf={tf}: probe {best['ms_probe_agg']:.3f} ms "
              f"orders {best['ms_orders_build']:.3f} ms "
              f"{'OK' if ok else 'PARITY MISMATCH!'}", flush=True)
        results[tf] = (best["ms_probe_agg"], best["ms_orders_build"])
        q.free()
    print(json.dumps(results))


if __name__ == "__main__":
    main()
