import pytest
import torch

from torchacc_amd.config import Config, DistConfig
from torchacc_amd.dist.mesh import ProcessTopology
from torchacc_amd.utils.utils import (partition_balanced, partition_uniform)


def test_config_defaults_validate():
    c = Config()
    c.compute.validate()
    c.memory.validate()
    c.dataloader.validate()
    c.dist.validate(world_size=1)


def test_fp16_bf16_exclusive():
    c = Config()
    c.compute.fp16 = True
    c.compute.bf16 = True
    with pytest.raises(AssertionError):
        c.compute.validate()


def test_dp_inference():
    d = DistConfig()
    d.fsdp.size = 2
    d.validate(world_size=8)
    assert d.dp.size == 4


def test_world_size_mismatch():
    d = DistConfig()
    d.fsdp.size = 3
    with pytest.raises(ValueError):
        d.validate(world_size=8)


def test_topology_coords():
    t = ProcessTopology(["dp", "tp"], [2, 4])
    assert t.world_size() == 8
    assert t.get_rank(dp=1, tp=2) == 6
    assert t.get_axis_rank(6, "dp") == 1
    assert t.get_axis_rank(6, "tp") == 2
    lists = t.get_axis_comm_lists("tp")
    assert lists == [[0, 1, 2, 3], [4, 5, 6, 7]]
    lists = t.get_axis_comm_lists("dp")
    assert lists == [[0, 4], [1, 5], [2, 6], [3, 7]]


def test_partition_uniform():
    assert partition_uniform(10, 3) == [0, 4, 7, 10]
    assert partition_uniform(4, 4) == [0, 1, 2, 3, 4]


def test_partition_balanced():
    w = [1, 1, 1, 10, 1, 1]
    bounds = partition_balanced(w, 2)
    assert bounds[0] == 0 and bounds[-1] == 6
    # heavy item isolated reasonably: max part weight minimized
    parts = [sum(w[bounds[i]:bounds[i + 1]]) for i in range(2)]
    assert max(parts) <= 12
