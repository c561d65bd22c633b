"""Create-time validation across every operator family: bad configs must
fail with a NULL handle + gx_last_error message (the int-code +
lastError contract INTEGRATION.md maps to TddlRuntimeException), never
crash or return a half-built op. Oracle build; the HIP library shares
the same validation code paths (identical create-layer source)."""
import numpy as np
import pytest

from galaxysql_amd import abi
from galaxysql_amd.chunk import I64, F64
from galaxysql_amd.operators import (EquiJoinKey, ParallelHashJoinExec,
                                     HashAggExec, OverWindowFramesExec,
                                     NonFrameOverWindowExec, ScanExec,
                                     PartitioningExchanger)


@pytest.fixture(scope="module")
def lib():
    import os
    import subprocess
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    subprocess.run(["make", "-C", os.path.join(repo, "oracle")], check=True,
                   capture_output=True)
    return abi.load_oracle()


def test_join_rejects_bad_type(lib):
    with pytest.raises(RuntimeError):
        ParallelHashJoinExec(lib, 99, [EquiJoinKey(0, 0, I64)],
                             [I64], [I64], device=-1)


def test_agg_rejects_window_only_func(lib):
    with pytest.raises(RuntimeError):
        HashAggExec(lib, group_cols=[0], aggs=[(abi.RANK, -1)],
                    input_types=[I64], device=-1)


def test_agg_rejects_unknown_func(lib):
    with pytest.raises(RuntimeError):
        HashAggExec(lib, group_cols=[0], aggs=[(123, 1)],
                    input_types=[I64, I64], device=-1)


def test_window_rejects_unknown_func(lib):
    with pytest.raises(RuntimeError):
        NonFrameOverWindowExec(lib, [0], [(123, 1)], [I64, I64], device=-1)


def test_fwindow_rejects_nonadditive_range(lib):
    # RANGE frames need a numeric order column: SLICE order col rejected
    from galaxysql_amd.chunk import SLICE
    with pytest.raises(RuntimeError):
        OverWindowFramesExec(
            lib, [0],
            [(abi.SUM_I64, 1, abi.FRAME_RANGE_SLIDING, 5, 5, 2, 1)],
            [I64, I64, SLICE], device=-1)


def test_scan_rejects_bad_proj_col(lib):
    with pytest.raises(RuntimeError):
        ScanExec(lib, [], [(abi.PROJ_COPY, 7, -1)], [I64], device=-1)


def test_error_message_is_nonempty(lib):
    try:
        HashAggExec(lib, group_cols=[0], aggs=[(123, 1)],
                    input_types=[I64, I64], device=-1)
    except RuntimeError as e:
        assert str(e).strip(), "gx_last_error must carry a message"
    else:
        pytest.fail("expected create failure")
