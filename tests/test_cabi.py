"""CPU-side checks of the C-ABI library: it builds, loads, and exports every
symbol include/distributed_join.h declares. No compute calls (no GPU here).
"""
import ctypes
import os
import re
import subprocess

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HEADER = os.path.join(REPO, "include", "distributed_join.h")
SO = os.path.join(REPO, "distributed_join_amd", "libdistjoin.so")


def _build():
    if not os.path.exists(SO):
        subprocess.run(["make", "-C", os.path.join(REPO, "distributed_join_amd", "csrc")],
                       check=True, capture_output=True)


def _declared_symbols():
    src = open(HEADER).read()
    # function declarations: return type then dj_name(
    return sorted(set(re.findall(r"\b(dj_\w+)\s*\(", src)))


def test_library_loads_and_exports_all_header_symbols():
    _build()
    lib = ctypes.CDLL(SO)
    syms = _declared_symbols()
    assert len(syms) >= 25
    missing = [s for s in syms if not hasattr(lib, s)]
    assert not missing, f"missing exports: {missing}"


def test_device_count_callable_without_gpu():
    _build()
    import distributed_join_amd as dj
    n = dj.lib().dj_device_count()
    assert n >= 0  # 0 in this container


def test_product_path_fails_loudly_without_gpu():
    _build()
    import distributed_join_amd as dj
    if dj.lib().dj_device_count() == 0:
        with pytest.raises(dj.ExtensionMissing):
            dj.require_gpu()
