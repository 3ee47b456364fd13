"""The C-ABI boundary check the scope table requires: both shared
libraries load (no GPU needed — loading links, it does not launch), and
every function include/*.h declares resolves to an exported symbol."""
import ctypes
import os
import re

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

HEADERS = {
    "distmlip_graph.h": "libdistmlip_graph.so",
    "distmlip_hip.h": "libdistmlip_hip.so",
}


def _declared(header_path):
    src = open(header_path).read()
    src = re.sub(r"/\*.*?\*/", "", src, flags=re.S)     # strip comments
    src = re.sub(r"//[^\n]*", "", src)
    return sorted(set(re.findall(r"\b(dm_\w+)\s*\(", src)))


@pytest.mark.parametrize("header,libname", sorted(HEADERS.items()))
def test_every_declared_symbol_exports(header, libname):
    from distmlip_amd.capi import _load
    lib = _load(libname)
    syms = _declared(os.path.join(ROOT, "include", header))
    assert syms, f"no declarations parsed from {header}"
    missing = [s for s in syms if not hasattr(lib, s)]
    assert not missing, f"{libname} missing exports: {missing}"


def test_ctypes_loads_without_package():
    """The .so is consumable by plain ctypes (the FFI story in
    INTEGRATION.md) — no Python package required."""
    for libname in HEADERS.values():
        path = os.path.join(ROOT, "distmlip_amd", "csrc", libname)
        if not os.path.exists(path):
            pytest.skip(f"{libname} not built")
        lib = ctypes.CDLL(path)
        assert lib is not None


def test_bench_cpu_baseline_leg_runs():
    """The bench contract's cpu_baseline leg (oracle timed on host cores)
    must stay runnable without a GPU and report the declared fields."""
    import bench
    out = bench.cpu_baseline_leg("li100k", 4)
    assert out["kind"] == "port"
    assert out["unit"] == "atom_steps_per_s"
    assert out["value"] > 0
    assert out["cores"] == 4
    assert "sample" in out
