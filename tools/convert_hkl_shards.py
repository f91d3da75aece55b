#!/usr/bin/env python3
"""Convert reference-format hickle image shards to this framework's
.npy shard format (the reference stored ImageNet as
hickle `.hkl` files of 256 images x 3x256x256 uint8, labels in .npy).

    python tools/convert_hkl_shards.py SRC_DIR DST_DIR [--labels DIR]

Requires `hickle` (not bundled; install wherever the .hkl data lives).
Output: `<name>.npy` as uint8 [n, H, W, 3] (NHWC) plus
`<name>_labels.npy`, directly consumable by
theanompi_amd.models.data.imagenet.ImageNetData.  Weight checkpoints
need no conversion — both sides use per-parameter .npy
(W_<i>_<epoch>.npy / V_<i>_<epoch>.npy).
"""

import argparse
import glob
import os
import sys

import numpy as np


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("src")
    ap.add_argument("dst")
    ap.add_argument("--labels", default=None,
                    help="directory of per-shard label .npy files "
                         "(default: alongside the .hkl files)")
    args = ap.parse_args()
    try:
        import hickle
    except ImportError:
        sys.exit("hickle is not installed; run this where the "
                 "reference dataset lives (`pip install hickle`)")
    os.makedirs(args.dst, exist_ok=True)
    shards = sorted(glob.glob(os.path.join(args.src, "*.hkl")))
    if not shards:
        sys.exit(f"no .hkl files in {args.src}")
    for path in shards:
        name = os.path.splitext(os.path.basename(path))[0]
        arr = np.asarray(hickle.load(path))
        # reference layout: [3, H, W, n] (c01b) or [n, 3, H, W]
        if arr.ndim != 4:
            sys.exit(f"{path}: expected 4-D array, got {arr.shape}")
        if arr.shape[0] == 3:                      # c01b -> NHWC
            arr = arr.transpose(3, 1, 2, 0)
        elif arr.shape[1] == 3:                    # NCHW -> NHWC
            arr = arr.transpose(0, 2, 3, 1)
        arr = np.ascontiguousarray(arr.astype(np.uint8))
        np.save(os.path.join(args.dst, name + ".npy"), arr)
        ldir = args.labels or args.src
        for cand in (name + "_labels.npy", name + ".labels.npy",
                     "labels_" + name + ".npy"):
            lp = os.path.join(ldir, cand)
            if os.path.exists(lp):
                lab = np.load(lp).astype(np.int64)
                np.save(os.path.join(args.dst, name + "_labels.npy"),
                        lab)
                break
        else:
            print(f"warning: no labels found for {name}", file=sys.stderr)
        print(f"{name}: {arr.shape} -> ok")


if __name__ == "__main__":
    main()
