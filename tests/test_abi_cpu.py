"""CPU-side checks of the C-ABI library: it builds, loads, and exports every
symbol include/gpuexec.h declares.  No compute calls (no GPU here)."""
import ctypes
import os
import re

import pytest

HERE = os.path.dirname(os.path.abspath(__file__))
ROOT = os.path.dirname(HERE)
SO = os.path.join(ROOT, "cloudberry_amd", "libgpuexec.so")
HDR = os.path.join(ROOT, "include", "gpuexec.h")


def _build_if_needed():
    if not os.path.exists(SO):
        import __graft_entry__
        __graft_entry__.build()


def test_library_builds_and_loads():
    _build_if_needed()
    lib = ctypes.CDLL(SO)
    lib.gx_version.restype = ctypes.c_char_p
    assert b"gfx950" in lib.gx_version()


def test_every_declared_symbol_exported():
    _build_if_needed()
    lib = ctypes.CDLL(SO)
    with open(HDR) as f:
        hdr = f.read()
    # every gx_* function declared in the header
    names = set(re.findall(r"\b(gx_[a-z0-9_]+)\s*\(", hdr))
    names -= {n for n in names if n in ("gx_status",)}
    assert len(names) >= 15
    for n in sorted(names):
        assert hasattr(lib, n), f"symbol {n} not exported"


def test_init_without_gpu_fails_loudly():
    """On a GPU-less host gx_init must return GX_ERR_NOGPU — never a silent
    CPU fallback (the round-end 'native code not loaded' check)."""
    try:
        import torch
        if torch.cuda.is_available():
            pytest.skip("GPU present; covered by gpu tests")
    except ImportError:
        pass
    _build_if_needed()
    lib = ctypes.CDLL(SO)
    lib.gx_init.argtypes = [ctypes.c_int, ctypes.c_int, ctypes.c_int,
                            ctypes.POINTER(ctypes.c_void_p)]
    h = ctypes.c_void_p()
    st = lib.gx_init(0, 0, 1, ctypes.byref(h))
    assert st == 6  # GX_ERR_NOGPU


def test_division_free_addressing_exact():
    """gx_mulhi64(row, magic) must equal row // rpb for every width's rpb
    (the GPU scan kernels rely on it for O(1) AOCS addressing)."""
    _build_if_needed()
    lib = ctypes.CDLL(SO)
    lib.gx_selftest_addressing.restype = ctypes.c_int
    assert lib.gx_selftest_addressing() == 0


def test_gfx950_code_object_embedded():
    _build_if_needed()
    with open(SO, "rb") as f:
        blob = f.read()
    assert b"gfx950" in blob


def test_customscan_extension_compiles():
    """VERDICT r01 missing #1: integration/gpuexec_cb.c must compile to an
    object against the reference server headers (stub pg_config + generated
    lwlocknames/fmgroids/errcodes/catalog headers), and its local mirror of
    the gx_* ABI structs must match include/gpuexec.h (abi-check compiles
    the file WITH the real header).  Skipped where the reference tree is
    absent (GPU boxes)."""
    import subprocess
    root = os.path.dirname(HERE)
    if not os.path.isdir("/root/reference/src/include"):
        pytest.skip("reference tree absent")
    subprocess.run(["make", "-C", os.path.join(root, "integration"), "clean"],
                   check=True, capture_output=True)
    r = subprocess.run(["make", "-C", os.path.join(root, "integration")],
                       capture_output=True, text=True)
    assert r.returncode == 0, r.stdout + r.stderr
    assert os.path.exists(os.path.join(root, "integration", "gpuexec_cb.o"))
    r = subprocess.run(["make", "-C", os.path.join(root, "integration"),
                        "abi-check"], capture_output=True, text=True)
    assert r.returncode == 0, r.stdout + r.stderr


def test_ctypes_struct_sizes_match_c():
    """Pin the ctypes mirrors to the C structs (an r2 mid-struct insertion
    once shifted every int64 stat field — this test makes that impossible
    to repeat silently)."""
    import cloudberry_amd as gx_pkg
    lib = gx_pkg.lib()
    lib.gx_abi_sizeof.restype = ctypes.c_int64
    lib.gx_abi_sizeof.argtypes = [ctypes.c_int]
    assert lib.gx_abi_sizeof(0) == ctypes.sizeof(gx_pkg._Stats)
    assert lib.gx_abi_sizeof(1) == ctypes.sizeof(gx_pkg._Group)
    assert lib.gx_abi_sizeof(2) == ctypes.sizeof(gx_pkg._KvGroup)
    assert lib.gx_abi_sizeof(3) == ctypes.sizeof(gx_pkg._Q3Desc)
    assert lib.gx_abi_sizeof(4) == ctypes.sizeof(gx_pkg._ColDesc)
    assert lib.gx_abi_sizeof(5) == ctypes.sizeof(gx_pkg._Filter)
