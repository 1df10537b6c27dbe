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
