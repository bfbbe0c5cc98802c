"""Write MNIST-shaped training files in the reference's on-disk format.

Analog of the reference's download_dataset.py:9-29 (sklearn
fetch_openml("mnist_784") → normalize → one-hot → 85/15 split →
x_{train,val}.parquet + y_{train,val}.npy).  This environment has no
network, so by default the data is SYNTHETIC MNIST-shaped (random
linear teacher — learnable); if a real mnist npz/csv is available it
could be dropped in the same format.

Usage: python prepare_data.py [--out data] [--samples 70000]
"""

import argparse
import os

import numpy as np
import pandas as pd
import torch

from shallowspeed_amd.data import synthesize


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", default="data")
    ap.add_argument("--samples", type=int, default=70000)
    ap.add_argument("--seed", type=int, default=42)
    args = ap.parse_args()

    x, y = synthesize(args.samples, 784, 10, seed=args.seed)
    # normalize like the reference (x/255 then mean-center is moot for
    # synthetic N(0,1) data, but keep the mean-centering step)
    x = x - x.mean()

    # 85/15 split (reference download_dataset.py:19-21, seed 42)
    g = torch.Generator().manual_seed(args.seed)
    perm = torch.randperm(args.samples, generator=g)
    n_train = int(args.samples * 0.85)
    idx = {"train": perm[:n_train], "val": perm[n_train:]}

    os.makedirs(args.out, exist_ok=True)
    for split, ii in idx.items():
        xs = x[ii].numpy()
        ys = y[ii].numpy().astype(np.float32)
        pd.DataFrame(xs).to_parquet(os.path.join(args.out, f"x_{split}.parquet"))
        np.save(os.path.join(args.out, f"y_{split}.npy"), ys)
        print(f"{split}: x{xs.shape} y{ys.shape}")
    print(f"wrote {args.out}/x_{{train,val}}.parquet + y_{{train,val}}.npy")


if __name__ == "__main__":
    main()
