"""Convert a local HF checkpoint into the servable per-block npy layout.

Usage:  python -m bloombee_amd.cli.convert /path/to/hf-model [out_dir]

The output directory is directly servable and loadable:
    python -m bloombee_amd.cli.run_server <out_dir> --block-indices 0:16
    AutoDistributedModelForCausalLM.from_pretrained(<out_dir>, ...)
(reference flow: server/from_pretrained.py downloads + converts on demand;
this environment has no hub egress, so conversion is an explicit local step.)
"""
from __future__ import annotations

import argparse


def main() -> None:
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("hf_dir", help="local HF checkpoint directory")
    ap.add_argument("out_dir", nargs="?", default=None,
                    help="output dir (default: {hf_dir}-np)")
    ap.add_argument("--force", action="store_true",
                    help="re-convert even if the sentinel exists")
    args = ap.parse_args()

    from bloombee_amd.server.from_pretrained import convert_hf_checkpoint

    out = convert_hf_checkpoint(args.hf_dir, args.out_dir, force=args.force)
    print(out)


if __name__ == "__main__":
    main()
