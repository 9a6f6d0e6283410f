"""Exact rank-layout assertions per mode (mirrors reference test strategy,
tests/distributed/_initializers/test_initialize_*_group.py)."""
import pytest

from pipegoose_amd.distributed.parallel_mode import ParallelMode
from pipegoose_amd.testing import init_parallel_context, spawn


def run_layouts(rank, world_size, port, tp, pp, dp, expected):
    ctx = init_parallel_context(
        rank, world_size, port,
        tensor_parallel_size=tp, pipeline_parallel_size=pp, data_parallel_size=dp,
    )
    for mode_name, layout in expected.items():
        mode = ParallelMode[mode_name]
        got = ctx.get_ranks_in_group(mode)
        assert got in layout, f"rank {rank} mode {mode}: {got} not in {layout}"
        assert rank in got
        assert ctx.get_local_rank(mode) == got.index(rank)
        assert ctx.get_world_size(mode) == len(got)
    ctx.destroy()


@pytest.mark.parametrize(
    "tp,pp,dp,expected",
    [
        # world=8: tp2 x pp2 x dp2 — layout [pp][dp][tp]
        (2, 2, 2, {
            "TENSOR": [[0, 1], [2, 3], [4, 5], [6, 7]],
            "PIPELINE": [[0, 4], [1, 5], [2, 6], [3, 7]],
            "DATA": [[0, 2], [1, 3], [4, 6], [5, 7]],
            "EXPERT_DATA": [[0, 2], [1, 3], [4, 6], [5, 7]],
        }),
        (4, 1, 2, {
            "TENSOR": [[0, 1, 2, 3], [4, 5, 6, 7]],
            "PIPELINE": [[0], [1], [2], [3], [4], [5], [6], [7]],
            "DATA": [[0, 4], [1, 5], [2, 6], [3, 7]],
        }),
    ],
)
def test_group_layouts(tp, pp, dp, expected):
    spawn(run_layouts, world_size=tp * pp * dp, tp=tp, pp=pp, dp=dp, expected=expected)


def run_single(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port)
    for mode in ParallelMode:
        assert ctx.get_world_size(mode) == 1
        assert ctx.get_local_rank(mode) == 0
    assert ctx.get_next_global_rank() == 0
    ctx.destroy()


def test_world_size_one():
    spawn(run_single, world_size=1)
