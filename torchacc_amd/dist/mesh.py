"""Process mesh / named-axis topology.

Re-implements the semantics of the reference's torchacc/dist/mesh.py
(ProcessTopology :13-222, Mesh :225-419): a cartesian rank<->coordinate map
over named axes, plus per-axis torch.distributed process groups. Pure Python,
backend-agnostic (gloo for CPU tests, RCCL on MI355X).
"""
import itertools
from collections import namedtuple
from typing import Dict, List, Optional, Sequence

import torch.distributed as dist

from ..utils.logger import logger


class ProcessTopology:
    """Maps global ranks <-> coordinates in a named-axis cartesian grid.

    Axes are ordered outermost-first: the FIRST axis has the LARGEST rank
    stride (spans nodes), the LAST axis indexes adjacent ranks (intra-node,
    over xGMI). Example: axes=['dp','tp'], dims=[2,4] ->
    rank = dp*4 + tp.
    """

    def __init__(self, axes: Sequence[str], dims: Sequence[int]):
        assert len(axes) == len(dims)
        self.axes = list(axes)
        self.dims = list(dims)
        self.ProcessCoord = namedtuple("ProcessCoord", self.axes)
        self._coord_to_rank: Dict[tuple, int] = {}
        self._rank_to_coord: List[tuple] = []
        for rank, coord in enumerate(itertools.product(
                *[range(d) for d in self.dims])):
            c = self.ProcessCoord(*coord)
            self._coord_to_rank[c] = rank
            self._rank_to_coord.append(c)

    def world_size(self) -> int:
        n = 1
        for d in self.dims:
            n *= d
        return n

    def get_dim(self, axis: str) -> int:
        return self.dims[self.axes.index(axis)]

    def get_rank(self, **coord_kw) -> int:
        assert sorted(coord_kw.keys()) == sorted(self.axes), \
            f"need all axes {self.axes}, got {list(coord_kw)}"
        return self._coord_to_rank[self.ProcessCoord(**coord_kw)]

    def get_coord(self, rank: int):
        return self._rank_to_coord[rank]

    def get_axis_rank(self, rank: int, axis: str) -> int:
        return getattr(self.get_coord(rank), axis)

    def filter_match(self, **filter_kw) -> List[int]:
        """All ranks whose coordinate matches the given axis values."""
        out = []
        for rank, coord in enumerate(self._rank_to_coord):
            if all(getattr(coord, k) == v for k, v in filter_kw.items()):
                out.append(rank)
        return out

    def get_axis_comm_lists(self, axis: str) -> List[List[int]]:
        """Rank lists for communication along ``axis``: one list per
        combination of the other axes' coordinates; each list varies only
        along ``axis`` (reference mesh.py:133)."""
        if axis not in self.axes:
            return []
        other_axes = [a for a in self.axes if a != axis]
        lists = []
        for combo in itertools.product(
                *[range(self.get_dim(a)) for a in other_axes]):
            fixed = dict(zip(other_axes, combo))
            ranks = [
                self.get_rank(**{axis: i}, **fixed)
                for i in range(self.get_dim(axis))
            ]
            lists.append(ranks)
        return lists

    def __str__(self):
        return f"ProcessTopology(axes={self.axes}, dims={self.dims})"


class Mesh:
    """Builds and caches per-axis process groups from a topology ordering.

    Axis sizes of 1 are kept in the topology (size-1 groups are created with
    ``use_local_synchronization``-free new_group over single ranks only when
    needed — we skip group creation for singleton axes and return None,
    callers treat None as "no communication").
    """

    _AXES = ("dp", "fsdp", "pp", "tp")

    def __init__(self,
                 dp_num: int = 1,
                 pp_num: int = 1,
                 tp_num: int = 1,
                 fsdp_num: int = 1,
                 topology: Optional[List[str]] = None):
        if not dist.is_initialized():
            raise RuntimeError("torch.distributed must be initialized "
                               "before building a Mesh")
        world = dist.get_world_size()
        sizes = {"dp": dp_num, "fsdp": fsdp_num, "pp": pp_num, "tp": tp_num}
        prod = dp_num * fsdp_num * pp_num * tp_num
        assert prod == world, \
            f"dp*fsdp*pp*tp = {prod} != world_size {world}"
        topology = list(topology or ["dp", "fsdp", "pp", "tp"])
        assert sorted(topology) == sorted(self._AXES)
        self.topology = ProcessTopology(topology,
                                        [sizes[a] for a in topology])
        self.global_rank = dist.get_rank()
        self._groups: Dict[str, Optional[dist.ProcessGroup]] = {}
        self._rank_groups: Dict[str, List[List[int]]] = {}
        # Group creation is collective: every rank must create every group.
        for axis in topology:
            self._build_axis_groups(axis)
        if self.global_rank == 0:
            logger.info("Mesh: %s", self.topology)

    def _build_axis_groups(self, axis: str):
        lists = self.topology.get_axis_comm_lists(axis)
        self._rank_groups[axis] = lists
        if self.topology.get_dim(axis) == 1 or len(lists) <= 0:
            self._groups[axis] = None
            return
        if len(lists) == 1 and len(lists[0]) == dist.get_world_size():
            # whole-world axis: reuse the default group
            self._groups[axis] = dist.group.WORLD
            return
        my_group = None
        for ranks in lists:
            g = dist.new_group(ranks=ranks)
            if self.global_rank in ranks:
                my_group = g
        self._groups[axis] = my_group

    # ---- generic accessors ---------------------------------------------

    def _axis_rank(self, axis: str) -> int:
        return self.topology.get_axis_rank(self.global_rank, axis)

    def _axis_num(self, axis: str) -> int:
        return self.topology.get_dim(axis)

    def get_dp_rank(self):
        return self._axis_rank("dp")

    def get_dp_num(self):
        return self._axis_num("dp")

    def get_dp_proc_group(self):
        return self._groups["dp"]

    def get_dp_rank_groups(self):
        return self._rank_groups["dp"]

    def get_fsdp_rank(self):
        return self._axis_rank("fsdp")

    def get_fsdp_num(self):
        return self._axis_num("fsdp")

    def get_fsdp_proc_group(self):
        return self._groups["fsdp"]

    def get_fsdp_rank_groups(self):
        return self._rank_groups["fsdp"]

    def get_tp_rank(self):
        return self._axis_rank("tp")

    def get_tp_num(self):
        return self._axis_num("tp")

    def get_tp_proc_group(self):
        return self._groups["tp"]

    def get_tp_rank_groups(self):
        return self._rank_groups["tp"]

    def get_pp_rank(self):
        return self._axis_rank("pp")

    def get_pp_num(self):
        return self._axis_num("pp")

    def get_pp_proc_group(self):
        return self._groups["pp"]

    def get_pp_rank_groups(self):
        return self._rank_groups["pp"]

    # ---- PP helpers (reference mesh.py:351-365) -------------------------

    def get_stage_id(self) -> int:
        return self._axis_rank("pp")

    def is_first_stage(self) -> bool:
        return self.get_stage_id() == 0

    def is_last_stage(self) -> bool:
        return self.get_stage_id() == self._axis_num("pp") - 1

    def stage_to_global(self, stage_id: int) -> int:
        """Global rank of the process at PP stage ``stage_id`` sharing all
        this rank's other axis coordinates."""
        coord = self.topology.get_coord(self.global_rank)
        kw = coord._asdict()
        kw["pp"] = stage_id
        return self.topology.get_rank(**kw)
