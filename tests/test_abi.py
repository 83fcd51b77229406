"""The C-ABI library loads and exports every entry point include/gemx.h
declares. No compute calls — runs without a GPU (gemx only refuses work,
not loading, when no HIP device is present)."""
import ctypes
import os
import re

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HDR = os.path.join(REPO, "include", "gemx.h")
SO = os.path.join(REPO, "opengemini_amd", "libgemx.so")


def declared_functions():
    src = open(HDR).read()
    # strip comments, then take identifiers declared as functions:
    #   <ret> gemx_<name>(...)
    src = re.sub(r"/\*.*?\*/", "", src, flags=re.S)
    names = re.findall(r"\b(gemx_[a-z0-9_]+)\s*\(", src)
    # keep declarations only (each appears once in the cleaned header)
    return sorted(set(names))


def test_header_declares_expected_surface():
    names = declared_functions()
    for must in [
        "gemx_abi_version", "gemx_last_error", "gemx_device_count",
        "gemx_shard_attach", "gemx_shard_close",
        "gemx_scan_agg", "gemx_scan_agg_grouped", "gemx_scan_agg_ex",
        "gemx_preagg_build", "gemx_scan_preagg",
        "gemx_prom_rate", "gemx_prom_irate", "gemx_prom_over_time",
    ]:
        assert must in names, f"{must} missing from include/gemx.h"


def test_library_exports_every_declared_symbol():
    if not os.path.exists(SO):
        pytest.skip("libgemx.so not built in this checkout")
    lib = ctypes.CDLL(SO)
    missing = [n for n in declared_functions() if not hasattr(lib, n)]
    assert not missing, f"libgemx.so lacks declared symbols: {missing}"


def test_abi_version_and_no_silent_cpu_path():
    if not os.path.exists(SO):
        pytest.skip("libgemx.so not built in this checkout")
    lib = ctypes.CDLL(SO)
    lib.gemx_abi_version.restype = ctypes.c_int
    assert lib.gemx_abi_version() == 1
    # without a GPU, attach must refuse (GEMX_E_NOGPU) — never fall back
    lib.gemx_device_count.restype = ctypes.c_int
    if lib.gemx_device_count() == 0:
        lib.gemx_shard_attach.restype = ctypes.c_int
        h = ctypes.c_void_p()
        rc = lib.gemx_shard_attach(
            0, None, 0, None, 0, 3, ctypes.byref(h)
        )
        assert rc == -1  # GEMX_E_NOGPU
