"""Memory timeline / peak report (reference tools/plot_mem.py parsed XLA
buffer-assignment dumps; the eager equivalent reads torch.cuda memory
snapshots).

Two modes:
  python tools/plot_mem.py --snapshot mem.pickle     # offline snapshot file
  (in code)  from tools.plot_mem import record, report
"""
import argparse
import pickle
import sys
from collections import defaultdict


def record(path="mem.pickle"):
    """Call before the workload: enables history; dump with
    torch.cuda.memory._dump_snapshot(path) afterwards."""
    import torch
    torch.cuda.memory._record_memory_history(max_entries=200000)
    return path


def summarize_snapshot(snap) -> str:
    segs = snap.get("segments", snap) if isinstance(snap, dict) else snap
    by_stream = defaultdict(lambda: [0, 0])
    total = active = 0
    largest = []
    for seg in segs:
        total += seg.get("total_size", 0)
        st = seg.get("stream", 0)
        by_stream[st][0] += seg.get("total_size", 0)
        for b in seg.get("blocks", []):
            if b.get("state") == "active_allocated":
                active += b.get("size", 0)
                by_stream[st][1] += b.get("size", 0)
                largest.append((b.get("size", 0),
                                (b.get("frames") or [{}])[0].get(
                                    "name", "?")))
    largest.sort(reverse=True)
    lines = [f"reserved {total / 2**30:.2f} GiB, active "
             f"{active / 2**30:.2f} GiB"]
    for st, (tot, act) in sorted(by_stream.items()):
        lines.append(f"  stream {st}: reserved {tot / 2**30:.2f} GiB "
                     f"active {act / 2**30:.2f} GiB")
    lines.append("largest active blocks:")
    for size, name in largest[:15]:
        lines.append(f"  {size / 2**20:9.1f} MiB  {name}")
    return "\n".join(lines)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--snapshot", required=True,
                   help="pickle from torch.cuda.memory._dump_snapshot")
    args = p.parse_args()
    with open(args.snapshot, "rb") as f:
        snap = pickle.load(f)
    print(summarize_snapshot(snap))


if __name__ == "__main__":
    main()
