"""CLI: consolidate sharded FSDP checkpoints and optionally reshard them
(reference utils/consolidate_and_reshard_ckpts.py:12-157; installed as the
``consolidate_and_reshard_fsdp_ckpts`` console script)."""
import argparse

from ..dist.state_dict_utils import (MODEL_NAME_PATTERN, OPTIM_NAME_PATTERN,
                                     consolidate_and_reshard_fsdp_checkpoint)


def main(argv=None):
    p = argparse.ArgumentParser(
        description="Consolidate/reshard torchacc_amd FSDP checkpoints")
    p.add_argument("--ckpt_dir", required=True,
                   help="directory with rank-*-of-*-{model,optimizer}.pth")
    p.add_argument("--ckpt_name", default=MODEL_NAME_PATTERN,
                   help=f"model shard pattern (default {MODEL_NAME_PATTERN};"
                        f" optimizer uses {OPTIM_NAME_PATTERN})")
    p.add_argument("--ckpt_type", default="all",
                   choices=["all", "model", "optimizer"])
    p.add_argument("--reshard_num", type=int, default=0,
                   help="0 = consolidate only; N = also write N new shards")
    p.add_argument("--output_dir", default=None)
    args = p.parse_args(argv)
    out_dir = args.output_dir or args.ckpt_dir
    consolidate_and_reshard_fsdp_checkpoint(
        args.ckpt_dir, out_dir, args.reshard_num, args.ckpt_type)
    print(f"done: consolidated ({args.ckpt_type}) into {out_dir}" +
          (f", resharded to {args.reshard_num}" if args.reshard_num else ""))


if __name__ == "__main__":
    main()
