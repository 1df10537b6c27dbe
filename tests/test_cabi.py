"""C-ABI surface test (CPU): every function declared in include/sdb_gpu.h
must be exported by the product libraries; loading must not require a GPU.
No compute calls here (the GPU path fails loudly without a device — that
behavior itself is asserted)."""

import ctypes
import os
import re

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def declared_functions():
    hdr = open(os.path.join(REPO, "include", "sdb_gpu.h")).read()
    # int sdb_gpu_xxx(...) and const char* sdb_gpu_version(...)
    names = re.findall(r"^(?:int|const char\*)\s+(sdb_gpu_\w+)\s*\(", hdr,
                       re.M)
    assert len(names) >= 10, names
    return names


def test_gpu_lib_exports_every_declared_symbol():
    lib = ctypes.CDLL(os.path.join(REPO, "serenedb_amd", "libsdb_gpu.so"))
    for name in declared_functions():
        assert hasattr(lib, name), f"missing export: {name}"


def test_host_lib_exports():
    lib = ctypes.CDLL(os.path.join(REPO, "serenedb_amd", "libsdb_host.so"))
    for name in ("sdb_host_encode_doc_block", "sdb_host_decode_doc_block",
                 "sdb_host_build_segment", "sdb_host_build_synth_segment",
                 "sdb_host_segment_parse", "sdb_host_bm25_stats",
                 "sdb_host_topk_select"):
        assert hasattr(lib, name), name


def test_gpu_ctx_fails_loudly_without_gpu():
    import torch

    if torch.cuda.is_available():
        pytest.skip("GPU present")
    import serenedb_amd as sa

    with pytest.raises(RuntimeError, match="sdb_gpu_ctx_create"):
        sa.GpuContext(0)


def test_malformed_blob_rejected():
    """parse must reject garbage/truncated/wrong-version blobs loudly"""
    import numpy as np
    import serenedb_amd as sa

    class _View(ctypes.Structure):
        _fields_ = [("hdr", ctypes.c_void_p), ("terms", ctypes.c_void_p),
                    ("desc", ctypes.c_void_p), ("norms", ctypes.c_void_p),
                    ("payload", ctypes.c_void_p)]

    host = sa.host()
    v = _View()
    garbage = np.frombuffer(b"\x00" * 256, dtype=np.uint8)
    rc = host.sdb_host_segment_parse(
        garbage.ctypes.data_as(ctypes.c_void_p), ctypes.c_uint64(256),
        ctypes.byref(v))
    assert rc != 0
    # valid blob, truncated
    blob = sa.build_synth_segment(1, 1, 1000, [0.1])
    cut = np.frombuffer(blob[: len(blob) // 2], dtype=np.uint8)
    rc = host.sdb_host_segment_parse(
        cut.ctypes.data_as(ctypes.c_void_p), ctypes.c_uint64(len(cut)),
        ctypes.byref(v))
    assert rc != 0
    # wrong version
    bad = bytearray(blob[:256])
    bad[8] = 99
    arr = np.frombuffer(bytes(bad), dtype=np.uint8)
    rc = host.sdb_host_segment_parse(
        arr.ctypes.data_as(ctypes.c_void_p), ctypes.c_uint64(len(arr)),
        ctypes.byref(v))
    assert rc != 0
