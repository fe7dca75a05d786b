"""C-ABI surface tests (CPU-only): the product library must load, export
every symbol include/presto_gpu.h declares, and fail LOUDLY (not fall back)
when no GPU is present."""
import ctypes as C
import pathlib
import re

import pytest

REPO = pathlib.Path(__file__).resolve().parent.parent
SO = REPO / "presto_amd" / "libpresto_gpu.so"
HDR = REPO / "include" / "presto_gpu.h"


def _declared_functions():
    text = HDR.read_text()
    # function declarations: "type pg_xxx(...);"
    names = re.findall(r"\b(pg_[a-z0-9_]+)\s*\(", text)
    # drop struct/typedef matches; keep unique declaration names
    return sorted(set(names))


@pytest.fixture(scope="module")
def solib():
    if not SO.exists():
        import subprocess
        subprocess.run(["hipcc", "--offload-arch=gfx950", "-O3",
                        "-std=c++17", "-fPIC", "-ffp-contract=off", "-shared",
                        str(REPO / "presto_amd/csrc/kernels.hip"),
                        "-o", str(SO)], check=True)
    return C.CDLL(str(SO))


def test_all_header_symbols_exported(solib):
    missing = []
    for name in _declared_functions():
        try:
            getattr(solib, name)
        except AttributeError:
            missing.append(name)
    assert not missing, f"symbols declared but not exported: {missing}"


def test_no_gpu_fails_loudly(solib):
    """Without a GPU, creating an operator must error with a clear message
    — never silently fall back to CPU."""
    try:
        import torch
        if torch.cuda.is_available():
            pytest.skip("GPU present")
    except Exception:
        pass
    solib.pg_last_error.restype = C.c_char_p
    h = C.c_int64()
    # plan size intentionally valid-shaped: use a zeroed filter plan
    buf = C.create_string_buffer(16 + 8 * 24 + 4 + 16 * 16 + 256)
    st = solib.pg_op_create(1, buf, 0, C.byref(h))
    assert st != 0
    msg = (solib.pg_last_error() or b"").decode()
    assert "no AMD GPU" in msg or "plan size" in msg
    # with the real reason checked: device count is 0 here
    n = C.c_int32()
    solib.pg_device_count(C.byref(n))
    assert n.value == 0


def test_struct_sizes_match_ctypes(solib):
    """The ctypes mirrors in presto_amd/engine.py must match the C
    struct layouts exactly — a drift here corrupts every plan field
    after the divergence point."""
    from presto_amd import engine as E
    mirrors = [E.PgCol, E.PgPage, E.Pred, E.Proj, E.Agg,
               E.PlanFilterProject, E.PlanHashAggSmall, E.PlanHashBuild,
               E.PlanLookupJoin, E.PlanGroupBy, E.PlanTopN,
               E.PlanPartition]
    solib.pg_abi_struct_sizes.restype = C.c_int32
    n = solib.pg_abi_struct_sizes(None, 0)
    assert n == len(mirrors)
    sizes = (C.c_int32 * n)()
    solib.pg_abi_struct_sizes(sizes, n)
    for m, sz in zip(mirrors, sizes):
        assert C.sizeof(m) == sz, (m.__name__, C.sizeof(m), sz)
