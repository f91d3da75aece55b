"""AlexNet on an ImageNet-style shard pipeline — the reference's
flagship workflow (run scripts + hickle shards), on
the .npy shard loader with a separate loader process, per-epoch shard
reshuffle and dataset mean-image subtraction.

Point `TRAIN_DIR`/`VAL_DIR` at directories of `<name>.npy` image
shards (uint8 [n, H, W, 3]) with `<name>_labels.npy` alongside — or
run as-is to exercise the pipeline on generated synthetic shards.
"""

import glob
import os
import sys
import tempfile

TRAIN_DIR = os.environ.get("TRAIN_DIR", "")
VAL_DIR = os.environ.get("VAL_DIR", "")

if __name__ == "__main__":
    if not TRAIN_DIR:
        from theanompi_amd.models.data.imagenet import \
            write_synthetic_shards
        tmp = tempfile.mkdtemp(prefix="imagenet_demo_")
        paths = write_synthetic_shards(tmp, n_shards=8,
                                       images_per_shard=128, hw=256)
        TRAIN_DIR = VAL_DIR = tmp
        print(f"no TRAIN_DIR given - wrote synthetic shards to {tmp}")

    train = sorted(glob.glob(os.path.join(TRAIN_DIR, "*.npy")))
    train = [p for p in train if not p.endswith("_labels.npy")]
    val = sorted(glob.glob(os.path.join(VAL_DIR, "*.npy")))
    val = [p for p in val if not p.endswith("_labels.npy")]

    from theanompi_amd import BSP
    rule = BSP()
    rule.init(devices=["cuda0"],
              modelfile="theanompi_amd.models.alex_net",
              modelclass="AlexNet",
              epochs=2, sync_type="grad", exchange_strategy="rccl_bf16",
              train_shards=",".join(train), val_shards=",".join(val),
              save_dir="./ckpt_alexnet", save_freq=1)
    sys.exit(0 if all(c == 0 for c in rule.wait()) else 1)
