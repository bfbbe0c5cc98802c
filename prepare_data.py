"""Write MNIST-shaped training files in the reference's on-disk format.

Analog of the reference's download_dataset.py:9-29 (sklearn
fetch_openml("mnist_784") → normalize → one-hot → 85/15 split →
x_{train,val}.parquet + y_{train,val}.npy).  This environment has no
network, so two sources:

  * --source synthetic (default): MNIST-shaped random data labeled by
    a fixed linear teacher (learnable; convergence-testable).
  * --source digits: REAL handwritten digit images — sklearn's bundled
    load_digits() (1,797 8×8 grayscale digits, shipped with sklearn, no
    network needed), bilinearly upsampled to the reference's 28×28=784
    feature shape, /16 normalized then mean-centered (the reference
    normalizes /255 then mean-centers, download_dataset.py:14-16).
    This is the closest obtainable stand-in for fetch_openml MNIST in
    an offline environment: real images, real label noise, same tensor
    shapes and on-disk format end to end.

Usage: python prepare_data.py [--out data] [--samples 70000]
                              [--source synthetic|digits]
"""

import argparse
import os

import numpy as np
import pandas as pd
import torch

from shallowspeed_amd.data import synthesize


def load_real_digits():
    """Real 8×8 handwritten digits → MNIST-784-shaped (x, one_hot_y)."""
    from sklearn.datasets import load_digits

    d = load_digits()
    x8 = torch.from_numpy(d.data.astype(np.float32)).view(-1, 1, 8, 8) / 16.0
    x = torch.nn.functional.interpolate(
        x8, size=(28, 28), mode="bilinear", align_corners=False
    ).reshape(-1, 784)
    x = x - x.mean()
    labels = torch.from_numpy(d.target.astype(np.int64))
    y = torch.zeros(x.shape[0], 10, dtype=torch.float32)
    y[torch.arange(x.shape[0]), labels] = 1.0
    return x, y


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", default="data")
    ap.add_argument("--samples", type=int, default=70000)
    ap.add_argument("--seed", type=int, default=42)
    ap.add_argument("--source", default="synthetic",
                    choices=["synthetic", "digits"])
    args = ap.parse_args()

    if args.source == "digits":
        x, y = load_real_digits()
        args.samples = x.shape[0]
    else:
        x, y = synthesize(args.samples, 784, 10, seed=args.seed)
        # normalize like the reference (x/255 then mean-center is moot
        # for synthetic N(0,1) data, but keep the mean-centering step)
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
