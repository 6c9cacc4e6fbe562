"""CPU guards for the driver-facing harness files: they must stay
importable and their CLIs parseable — a syntax error in bench wiring
would otherwise only surface at round end on the GPU box."""
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_help_parses():
    r = subprocess.run([sys.executable, os.path.join(ROOT, "bench.py"),
                        "--help"], capture_output=True, text=True, timeout=60)
    assert r.returncode == 0
    for flag in ("--gpus", "--steps", "--warmup", "--sf", "--rle-keys",
                 "--force-motion", "--full-cpu-baseline", "--no-traffic"):
        assert flag in r.stdout


def test_bench_gpus_worldsize_assert():
    """--gpus N without N ranks must fail loudly."""
    env = dict(os.environ, WORLD_SIZE="1")
    r = subprocess.run([sys.executable, os.path.join(ROOT, "bench.py"),
                        "--gpus", "8", "--steps", "1"],
                       capture_output=True, text=True, timeout=60, env=env)
    assert r.returncode != 0
    assert "WORLD_SIZE" in (r.stderr + r.stdout)


def test_tool_modules_import():
    sys.path.insert(0, ROOT)
    import tools.pmc_traffic  # noqa: F401
    import __graft_entry__    # noqa: F401
    assert callable(tools.pmc_traffic.probe_kernel_fetch_bytes)
