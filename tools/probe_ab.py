#!/usr/bin/env python3
"""A/B the lineitem probe kernel variants at SF100 on one MI355X.

Round-2 candidates vs the shipped B=1 grid-stride kernel (variant 0):
  6  -> B=-1  block-chunked (contiguous table windows; L2 locality)
  14 -> B=-12 tile-compact-then-probe (r1: ~noise vs fused)
  16 -> B=-14 glds double-buffered tile scan (async global->LDS DMA)
Each timed run verifies hits/groups against variant 0 (parity gate).
"""
import json
import os
import sys
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)
import cloudberry_amd as gx  # noqa: E402

SF = float(os.environ.get("AB_SF", "100"))
REPS = int(os.environ.get("AB_REPS", "6"))


def run(q, variant, grid=None, tpb=None):
    os.environ["GX_PROBE_VARIANT"] = str(variant)
    if grid:
        os.environ["GX_PROBE_GRID"] = str(grid)
    elif "GX_PROBE_GRID" in os.environ:
        del os.environ["GX_PROBE_GRID"]
    if tpb:
        os.environ["GX_PROBE_TPB"] = str(tpb)
    elif "GX_PROBE_TPB" in os.environ:
        del os.environ["GX_PROBE_TPB"]
    best = None
    for _ in range(REPS):
        q.run()
        st = q.stats()
        if best is None or st["ms_probe_agg"] < best["ms_probe_agg"]:
            best = st
    return best


def main():
    ctx = gx.Context(device=0, seg=0, nsegs=1)
    t0 = time.time()
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, SF)
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, SF)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM, SF)
    q = ctx.q3(cust, ordr, li)
    print(f"setup {time.time()-t0:.1f}s", flush=True)

    base = run(q, 0)
    print(f"v0 (fused B=1, grid default): probe {base['ms_probe_agg']:.3f} ms "
          f"hits={base['probe_hits']} groups={base['groups']}", flush=True)

    results = {"v0": base["ms_probe_agg"]}
    for label, variant, grids in (
            ("v6_chunked", 6, (2048, 8192, 32768)),
            ("v14_tile", 14, (32768,)),
            ("v16_glds", 16, (2048, 4096, 8192, 16384, 32768, 65536)),
    ):
        for g in grids:
            st = run(q, variant, grid=g)
            ok = (st["probe_hits"] == base["probe_hits"] and
                  st["groups"] == base["groups"])
            print(f"{label} grid {g}: probe {st['ms_probe_agg']:.3f} ms "
                  f"{'OK' if ok else 'PARITY MISMATCH!'}", flush=True)
            results[f"{label}_g{g}"] = st["ms_probe_agg"]
            if not ok:
                results[f"{label}_g{g}_BAD"] = True
    os.makedirs(os.path.join(ROOT, "gpurun_out"), exist_ok=True)
    with open(os.path.join(ROOT, "gpurun_out", "probe_ab.json"), "w") as f:
        json.dump(results, f, indent=1)
    print(json.dumps(results))


if __name__ == "__main__":
    main()
