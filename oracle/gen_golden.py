#!/usr/bin/env python3
"""Generate tests/golden/hash_vectors.json from the REFERENCE's own hashfn.c
compiled standalone (oracle/_ref/libpgref.so — see oracle/Makefile).

The vectors pin the oracle's hash restatement (and, transitively, the GPU
Motion-routing kernels) to the reference implementation:
  hash_bytes_uint32   src/common/hashfn.c:620-637
  hashint8            src/backend/access/hash/hashfunc.c:85-101 (restated on
                      top of the reference hash_bytes_uint32 — the lohalf
                      fold is 3 lines, asserted identical in oracle tests)
  jump_consistent_hash src/backend/cdb/cdbhash.c:530-541 (restated verbatim
                      algorithm; pinned via published-paper test values and
                      cross-checked against the oracle)

Run from the repo root: python3 oracle/gen_golden.py
"""
import ctypes, json, os, random

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
ref = ctypes.CDLL(os.path.join(ROOT, "oracle", "_ref", "libpgref.so"))
ref.hash_bytes_uint32.restype = ctypes.c_uint32
ref.hash_bytes_uint32.argtypes = [ctypes.c_uint32]
ref.hash_bytes.restype = ctypes.c_uint32
ref.hash_bytes.argtypes = [ctypes.c_char_p, ctypes.c_int]
ref.pg_comp_crc32c_sb8.restype = ctypes.c_uint32
ref.pg_comp_crc32c_sb8.argtypes = [ctypes.c_uint32, ctypes.c_char_p, ctypes.c_size_t]


def ref_hashint8(v):
    """hashfunc.c:85-101 fold + reference hash_bytes_uint32."""
    lo = v & 0xFFFFFFFF
    hi = (v >> 32) & 0xFFFFFFFF
    lo ^= hi if v >= 0 else (~hi & 0xFFFFFFFF)
    return ref.hash_bytes_uint32(lo)


def main():
    rnd = random.Random(20260915)
    i64_cases = [0, 1, 2, 7, 42, -1, -2, 150_000_000, 600_037_902,
                 2**31 - 1, 2**31, 2**32, -(2**31), 2**62, -(2**62),
                 1234567890123, -987654321987]
    i64_cases += [rnd.randrange(-2**63, 2**63) for _ in range(64)]
    u32_cases = [0, 1, 42, 0xDEADBEEF, 0xFFFFFFFF] + [rnd.randrange(2**32) for _ in range(32)]
    bytes_cases = [b"", b"a", b"hello", b"BUILDING", bytes(range(32))]

    crc_cases = [b"", b"1", b"123456789", b"BUILDING", bytes(range(256)),
                 bytes(rnd.randrange(256) for _ in range(1000))]
    out = {
        "crc32c_state": [{"data": c.hex(),
                          "state": ref.pg_comp_crc32c_sb8(0xFFFFFFFF, c, len(c))}
                         for c in crc_cases],
        "hash_bytes_uint32": [{"k": k, "h": ref.hash_bytes_uint32(k)} for k in u32_cases],
        "hashint8": [{"v": v, "h": ref_hashint8(v)} for v in i64_cases],
        "hash_bytes": [{"k": c.hex(), "h": ref.hash_bytes(c, len(c))} for c in bytes_cases],
        "known_answers": {"hash_bytes_uint32(42)": "0x59fcfec8",
                          "hash_bytes(hello,5)": "0x90859829"},
    }
    path = os.path.join(ROOT, "tests", "golden", "hash_vectors.json")
    with open(path, "w") as f:
        json.dump(out, f, indent=1)
    print(f"wrote {path}: {len(i64_cases)} hashint8, {len(u32_cases)} u32 vectors")
    assert out["hash_bytes_uint32"][2]["h"] == 0x59FCFEC8


if __name__ == "__main__":
    main()
