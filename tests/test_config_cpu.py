"""Engine configuration parsing and registry hardening (CPU-only).

Covers the round-2 items: independent intra/cross reduction selection
(reference mpi_allreduce_operations.cc:90-115 — intra default SRA, cross
default Ring), the CGX_DEBUG_ALL_TO_ALL_REDUCTION debug flag, the
pointer-bound registry match (two buckets sharing a total numel but with
different layer layouts must not swap configs), and CGXState's explicit
error-feedback env reset.
"""

import os

import pytest

from torch_cgx_amd import _C


REDUCTION_VARS = ("CGX_REDUCTION_TYPE", "CGX_INNER_REDUCTION_TYPE",
                  "CGX_CROSS_REDUCTION_TYPE")


@pytest.fixture(autouse=True)
def _clean_env():
    saved = {k: os.environ.pop(k, None)
             for k in REDUCTION_VARS + ("CGX_DEBUG_ALL_TO_ALL_REDUCTION",
                                        "CGX_ERROR_FEEDBACK")}
    yield
    for k, v in saved.items():
        if v is None:
            os.environ.pop(k, None)
        else:
            os.environ[k] = v


def test_reduction_defaults():
    cfg = _C.parse_engine_config()
    assert cfg["inner_ring"] is False   # intra default: SRA
    assert cfg["cross_ring"] is True    # cross default: Ring


@pytest.mark.parametrize("inner", ["SRA", "Ring"])
@pytest.mark.parametrize("cross", ["SRA", "Ring"])
def test_reduction_matrix(inner, cross):
    os.environ["CGX_INNER_REDUCTION_TYPE"] = inner
    os.environ["CGX_CROSS_REDUCTION_TYPE"] = cross
    cfg = _C.parse_engine_config()
    assert cfg["inner_ring"] is (inner == "Ring")
    assert cfg["cross_ring"] is (cross == "Ring")


def test_reduction_legacy_alias_drives_both():
    os.environ["CGX_REDUCTION_TYPE"] = "Ring"
    cfg = _C.parse_engine_config()
    assert cfg["inner_ring"] is True and cfg["cross_ring"] is True
    os.environ["CGX_REDUCTION_TYPE"] = "SRA"
    cfg = _C.parse_engine_config()
    assert cfg["inner_ring"] is False and cfg["cross_ring"] is False
    # a specific variable overrides the alias
    os.environ["CGX_INNER_REDUCTION_TYPE"] = "Ring"
    cfg = _C.parse_engine_config()
    assert cfg["inner_ring"] is True and cfg["cross_ring"] is False


def test_debug_a2a_flag():
    assert _C.parse_engine_config()["debug_a2a"] is False
    os.environ["CGX_DEBUG_ALL_TO_ALL_REDUCTION"] = "1"
    assert _C.parse_engine_config()["debug_a2a"] is True


def test_registry_pointer_binding_disambiguates_equal_totals():
    # bucket 0: layers [100, 200]; bucket 1: layers [150, 150] — same total
    _C.clear_registry()
    try:
        _C.register_layer(0, 0, 100, 4, 128)
        _C.register_layer(0, 1, 200, 8, 256)
        _C.register_layer(1, 0, 150, 2, 64)
        _C.register_layer(1, 1, 150, 2, 64)
        ptr_a, ptr_b = 0x1000, 0x2000

        # first pass (registration order): learn the pointer bindings
        ok, idx = _C.registry_match(300, ptr_a)
        assert ok and idx == 0
        ok, idx = _C.registry_match(300, ptr_b)
        assert ok and idx == 1

        # second pass arrives in REVERSED order: the cursor heuristic alone
        # would hand bucket 0's config to ptr_b; pointer binding must not
        for _ in range(3):
            ok, idx = _C.registry_match(300, ptr_b)
            assert ok and idx == 1
            ok, idx = _C.registry_match(300, ptr_a)
            assert ok and idx == 0

        # a numel mismatch unbinds (bucket rebuild / storage reuse); the next
        # lookup falls back to the cursor heuristic (which bucket it picks is
        # the pre-existing ambiguity; what matters is it matches again)
        ok, _ = _C.registry_match(999, ptr_a)
        assert not ok
        ok, _ = _C.registry_match(300, ptr_a)
        assert ok
    finally:
        _C.clear_registry()


def test_registry_unknown_total_no_match():
    _C.clear_registry()
    try:
        _C.register_layer(0, 0, 64, 4, 128)
        ok, _ = _C.registry_match(65, 0x3000)
        assert not ok
    finally:
        _C.clear_registry()


def test_cgx_state_error_feedback_env_reset():
    import torch_cgx_amd

    torch_cgx_amd.CGXState(None, compression_params={"bits": 4,
                                                     "error_feedback": True})
    assert os.environ["CGX_ERROR_FEEDBACK"] == "1"
    # a later state WITHOUT error feedback must not inherit the stale "1"
    torch_cgx_amd.CGXState(None, compression_params={"bits": 4})
    assert os.environ["CGX_ERROR_FEEDBACK"] == "0"
    assert _C.parse_engine_config()["error_feedback"] is False
