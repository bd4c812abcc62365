#!/usr/bin/env python3
"""Pack an image dataset into RecordIO (parity with tools/im2rec.py).

Offline environment: sources are .npy arrays (H,W,C uint8/float) laid out
as root/<class>/<file>.npy; records are raw-encoded (MXARAW00) so no image
codec is needed.  Also generates the .lst file (index<TAB>label<TAB>path).
"""
import argparse
import os
import sys

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))
from anand_mxnet_amd import recordio


def make_list(root):
    items = []
    classes = sorted(d for d in os.listdir(root)
                     if os.path.isdir(os.path.join(root, d)))
    for label, cls in enumerate(classes):
        for fn in sorted(os.listdir(os.path.join(root, cls))):
            if fn.endswith(".npy"):
                items.append((label, os.path.join(cls, fn)))
    return items, classes


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("prefix", help="output prefix (prefix.rec/.idx/.lst)")
    ap.add_argument("root", help="dataset root (class folders of .npy)")
    ap.add_argument("--shuffle", action="store_true")
    args = ap.parse_args()

    items, classes = make_list(args.root)
    if args.shuffle:
        np.random.shuffle(items)
    with open(args.prefix + ".lst", "w") as f:
        for i, (label, rel) in enumerate(items):
            f.write(f"{i}\t{label}\t{rel}\n")
    rec = recordio.MXIndexedRecordIO(args.prefix + ".idx",
                                     args.prefix + ".rec", "w")
    for i, (label, rel) in enumerate(items):
        img = np.load(os.path.join(args.root, rel))
        if img.dtype != np.uint8:
            img = np.clip(img, 0, 255).astype(np.uint8)
        rec.write_idx(i, recordio.pack_img(
            recordio.IRHeader(0, float(label), i, 0), img))
    rec.close()
    print(f"wrote {len(items)} records, {len(classes)} classes -> "
          f"{args.prefix}.rec")


if __name__ == "__main__":
    main()
