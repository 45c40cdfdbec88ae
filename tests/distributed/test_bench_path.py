"""8-process rehearsal of the exact bench.py path the driver runs.

The driver launches ``torch.distributed.run --nproc-per-node 8 bench.py
--gpus 8`` on the GPU node; this test executes bench.main() itself at
world_size 8 over gloo (tiny model) so every wrapper/collective in that
path — config -> accelerate -> FSDP engine -> fused-optimizer step -> MAX
timing all-reduce -> JSON line — is proven multi-process before hardware.
"""
import contextlib
import io
import json
import os
import sys

from tests.utils.distributed import run_multiprocess

REPO_ROOT = os.path.dirname(os.path.dirname(
    os.path.dirname(os.path.abspath(__file__))))


def _bench_worker(rank, world):
    sys.path.insert(0, REPO_ROOT)
    sys.argv = ["bench.py", "--model", "tiny", "--steps", "2",
                "--warmup", "1", "--batch-size", "2", "--gpus", str(world)]
    import bench
    buf = io.StringIO()
    with contextlib.redirect_stdout(buf):
        bench.main()
    if rank == 0:
        line = buf.getvalue().strip().splitlines()[-1]
        out = json.loads(line)
        assert out["metric"] == "tokens/sec"
        assert out["n_gpus"] == world
        assert out["steps"] == 2 and out["warmup"] == 1
        assert out["value"] > 0
        assert out["config"]["parallelism"] == f"fsdp{world}"
        assert out["config"]["loss"] is not None


def test_bench_fsdp8_cpu_rehearsal():
    run_multiprocess(_bench_worker, world_size=8, timeout=600)


def _bench_cp_worker(rank, world, mode):
    sys.path.insert(0, REPO_ROOT)
    sys.argv = ["bench.py", "--model", "tiny", "--steps", "2",
                "--warmup", "1", "--batch-size", "2", "--mode", mode,
                "--seq-len", "256", "--gpus", str(world)]
    import bench
    buf = io.StringIO()
    with contextlib.redirect_stdout(buf):
        bench.main()
    if rank == 0:
        out = json.loads(buf.getvalue().strip().splitlines()[-1])
        assert out["value"] > 0
        assert out["config"]["parallelism"] == f"{mode}{world}"


def test_bench_ulysses4_cpu_rehearsal():
    run_multiprocess(_bench_cp_worker, world_size=4, args=("ulysses",),
                     timeout=600)


def test_bench_2d4_cpu_rehearsal():
    run_multiprocess(_bench_cp_worker, world_size=4, args=("2d",),
                     timeout=600)


def _bench_fsdp_tp_worker(rank, world):
    sys.path.insert(0, REPO_ROOT)
    sys.argv = ["bench.py", "--model", "tiny", "--steps", "2",
                "--warmup", "1", "--batch-size", "2", "--mode", "fsdp_tp",
                "--tp", "2", "--gpus", str(world)]
    import bench
    buf = io.StringIO()
    with contextlib.redirect_stdout(buf):
        bench.main()
    if rank == 0:
        out = json.loads(buf.getvalue().strip().splitlines()[-1])
        assert out["value"] > 0
        assert out["config"]["parallelism"] == "fsdp2xtp2"


def test_bench_fsdp_tp_cpu_rehearsal():
    """The 70B FSDPxTP launch path (BASELINE config 4) at tp2 x fsdp2."""
    run_multiprocess(_bench_fsdp_tp_worker, world_size=4, timeout=600)


def test_bench_ring2_cpu_rehearsal():
    run_multiprocess(_bench_cp_worker, world_size=2, args=("ring",),
                     timeout=600)
