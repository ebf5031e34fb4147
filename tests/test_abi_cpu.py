"""CPU-side checks of the C-ABI library: it loads without a GPU and exports
every symbol include/dingo_gpu.h declares.  No compute calls here (that
needs a device — tests/test_gpu_parity.py)."""
import ctypes as C
import os
import re

import pytest

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(HERE)
SO = os.path.join(REPO, "dingo-store_amd", "libdingo_gpu.so")
HDR = os.path.join(REPO, "include", "dingo_gpu.h")

pytestmark = pytest.mark.skipif(
    not os.path.exists(SO), reason="libdingo_gpu.so not built")


def header_symbols():
    src = open(HDR).read()
    src = re.sub(r"/\*.*?\*/", "", src, flags=re.S)
    src = re.sub(r"//[^\n]*", "", src)
    return sorted(set(re.findall(r"\b(dg_[a-z_0-9]+)\s*\(", src)))


def test_all_header_symbols_exported():
    lib = C.CDLL(SO)
    syms = header_symbols()
    assert len(syms) >= 15, syms
    missing = [s for s in syms if not hasattr(lib, s)]
    assert not missing, f"missing exports: {missing}"


def test_no_gpu_behavior():
    """Without a device the product path fails loudly (DG_ENOGPU), never
    silently falls back to CPU."""
    import torch
    if torch.cuda.is_available():
        pytest.skip("GPU present; covered by gpu tests")
    import sys
    sys.path.insert(0, os.path.join(REPO, "dingo-store_amd"))
    import dingostore as dg
    assert dg.device_count() == 0
    with pytest.raises(dg.DgError) as e:
        dg.Index(dg.FLAT, dg.L2, 64)
    assert "no HIP device" in str(e.value)


def test_build_info():
    import sys
    sys.path.insert(0, os.path.join(REPO, "dingo-store_amd"))
    import dingostore as dg
    assert "gfx950" in dg.build_info()


def test_mirror_selftest_symbol():
    lib = C.CDLL(SO)
    assert hasattr(lib, "dg_mirror_selftest")
