"""Checkpoint conversion CLI: torch snapshot <-> Caffe .caffemodel.

    python -m npairloss_amd.export --to-caffemodel snap_iter_100.pt out.caffemodel
    python -m npairloss_amd.export --to-pt net.caffemodel out.pt

Conversions go through the GoogLeNet Caffe layer-name map
(models/googlenet.py caffe_names); the resulting .caffemodel is readable
by Caffe itself (NetParameter V2 layers, name-matched blobs).
"""

from __future__ import annotations

import argparse

import torch


def main(argv=None):
    p = argparse.ArgumentParser()
    g = p.add_mutually_exclusive_group(required=True)
    g.add_argument("--to-caffemodel", action="store_true",
                   help="src is a trainer .pt snapshot (or state_dict); dst is .caffemodel")
    g.add_argument("--to-pt", action="store_true",
                   help="src is a .caffemodel; dst is a .pt state_dict")
    p.add_argument("src")
    p.add_argument("dst")
    p.add_argument("--model", default="googlenet")
    args = p.parse_args(argv)

    from .models import build_embedding_model
    from .utils.caffemodel import load_caffemodel_into, save_caffemodel

    net = build_embedding_model(args.model)
    if args.to_caffemodel:
        ck = torch.load(args.src, map_location="cpu", weights_only=False)
        state = ck["model"] if isinstance(ck, dict) and "model" in ck else ck
        net.load_state_dict(state)
        n = save_caffemodel(net, args.dst)
        print(f"wrote {args.dst}: {n} layers")
    else:
        loaded, skipped = load_caffemodel_into(net, args.src)
        torch.save(net.state_dict(), args.dst)
        print(f"wrote {args.dst}: loaded {len(loaded)} layers"
              + (f", skipped {len(skipped)} unmatched" if skipped else ""))


if __name__ == "__main__":
    main()
