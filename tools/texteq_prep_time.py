#!/usr/bin/env python3
"""Time the TEXT-predicate prepare (q3_build_text_mask) on a varlena
rle_type mktsegment column — the r1 path cost 510 ms at SF100 (15M rows);
r2 evaluates one texteq per RLE run directly over the AO blocks."""
import os
import sys
import time

import numpy as np

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)
import cloudberry_amd as gx  # noqa: E402
from oracle import pyapi as orc  # noqa: E402

NC = int(float(sys.argv[1])) if len(sys.argv) > 1 else 15_000_000

segs = [b"BUILDING", b"AUTOMOBILE", b"MACHINERY", b"HOUSEHOLD", b"FURNITURE"]
rng = np.random.default_rng(1)
idx = rng.integers(0, 5, NC)
t0 = time.time()
strings = [segs[i] for i in idx]
stream = orc.aocs_encode_varlena_rle(strings)
print(f"host encode {NC} rows: {time.time()-t0:.1f}s "
      f"({len(stream)/1e6:.1f} MB)", flush=True)

ctx = gx.Context(device=0, seg=0, nsegs=1)
cust = ctx.bind([(orc.aocs_encode(np.arange(1, NC + 1, dtype=np.int64)), 8, NC),
                 (stream, -1, NC, 1)])
ordr = ctx.tpch_gen(gx.TPCH_ORDERS, 0.1)
li = ctx.tpch_gen(gx.TPCH_LINEITEM, 0.1)
desc = {"dim": cust, "dim_key_col": 0, "dim_filter": (1, "==", "BUILDING"),
        "mid": ordr, "mid_key_col": 0, "mid_fk_col": 1,
        "mid_attr1_col": 2, "mid_attr2_col": 3,
        "mid_filter": (2, "<", gx.CUTOFF_19950315),
        "fact": li, "fact_key_col": 0, "fact_a_col": 1, "fact_b_col": 2,
        "fact_filter": (3, ">", gx.CUTOFF_19950315)}
t0 = time.perf_counter()
q = ctx.q3_desc(desc)
dt = (time.perf_counter() - t0) * 1000.0
print(f"q3_desc prepare (incl. per-run texteq mask) at {NC} rows: "
      f"{dt:.1f} ms", flush=True)
q.run()
print("run ok, groups:", len(q.result()["l_orderkey"]))
