"""Flat-parameter layout must agree between Python and the C++ extension."""
from parallel_cnn_amd import _C
from parallel_cnn_amd.ops import shapes as S


def test_offsets_match_extension():
    assert _C.N_PARAMS == S.N_PARAMS == 2343
    assert _C.OFF_C1W == S.OFF_C1W == 0
    assert _C.OFF_C1B == S.OFF_C1B == 150
    assert _C.OFF_S1W == S.OFF_S1W == 156
    assert _C.OFF_S1B == S.OFF_S1B == 172
    assert _C.OFF_FW == S.OFF_FW == 173
    assert _C.OFF_FB == S.OFF_FB == 2333


def test_reference_hyperparams():
    assert abs(_C.REF_DT - 0.1) < 1e-7  # fp32 literal
    assert abs(_C.REF_THRESHOLD - 1e-2) < 1e-7
