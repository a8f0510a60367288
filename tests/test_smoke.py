"""Import smoke tests (reference test/smoke_test.py): every public
subpackage imports, key entry points resolve."""
import importlib

import pytest


SUBPACKAGES = [
    "rl_amd",
    "rl_amd.tensordict",
    "rl_amd.data",
    "rl_amd.data.replay_buffers",
    "rl_amd.data.llm",
    "rl_amd.envs",
    "rl_amd.envs.transforms",
    "rl_amd.envs.libs",
    "rl_amd.modules",
    "rl_amd.objectives",
    "rl_amd.objectives.value",
    "rl_amd.collectors",
    "rl_amd.trainers",
    "rl_amd.record",
    "rl_amd.checkpoint",
    "rl_amd.parallel",
    "rl_amd.weight_update",
    "rl_amd.services",
    "rl_amd.render",
    "rl_amd.testing",
    "rl_amd.ops",
]


@pytest.mark.parametrize("name", SUBPACKAGES)
def test_imports(name):
    importlib.import_module(name)


def test_native_extension_present():
    from rl_amd import _C

    assert hasattr(_C, "COMPILED_WITH_HIP")
    assert hasattr(_C, "SumSegmentTreeFp32")


def test_entry_points():
    import __graft_entry__ as g

    assert callable(g.build) and callable(g.smoke)
    from rl_amd.render.cli import main

    assert callable(main)


def test_process_leak_helper():
    from rl_amd.testing.dist_utils import (
        assert_no_new_python_processes,
        snapshot_python_processes,
    )

    before = snapshot_python_processes()
    assert_no_new_python_processes(before)
