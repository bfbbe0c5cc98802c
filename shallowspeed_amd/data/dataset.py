"""Dataset: DP sharding + µbatch slicing, synthetic or file-backed.

Reference: shallowspeed/dataset.py:5-86 —
  * divisibility invariants global%DP==0, local%µbatch==0
    (dataset.py:35-39,60-61),
  * trim to a multiple of the global batch (dataset.py:50-52),
  * DP shard = strided slice [DP_rank : : DP_size] made contiguous
    (dataset.py:54-58),
  * µbatch row arithmetic start = batch_id*local + mubatch_id*µb
    (dataset.py:66-80).

There is no network in this environment, so the default is SYNTHETIC
MNIST-shaped data: x ~ N(0,1) (n,784) f32 and one-hot targets from a
fixed random linear teacher (so the task is learnable and convergence
is testable).  If the reference's files (x_train.parquet / y_train.npy,
written by its download_dataset.py:10-23) exist in save_dir they are
loaded instead.
"""

import os

import numpy as np
import torch


def synthesize(n_samples: int, in_dim: int = 784, n_classes: int = 10,
               seed: int = 1234, teacher_seed: int = 99991):
    """MNIST-shaped synthetic data labeled by a fixed random linear
    teacher.

    * The TEACHER is seeded independently of the split seed so train
      and validation splits are labeled by the same function
      (otherwise val accuracy is stuck at chance).
    * Samples are MARGIN-FILTERED: 2× candidates are generated and the
      half with the largest top1−top2 teacher-logit gap is kept
      (deterministically), so the task is cleanly learnable and
      convergence gates ("accuracy climbs", reference train.py:148-152)
      are crisp.
    """
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(2 * n_samples, in_dim, generator=g, dtype=torch.float32)
    gt = torch.Generator().manual_seed(teacher_seed + in_dim * 31 + n_classes)
    teacher = torch.randn(in_dim, n_classes, generator=gt, dtype=torch.float32)
    logits = x @ teacher
    top2 = logits.topk(min(2, n_classes), dim=1).values
    margin = top2[:, 0] - (top2[:, 1] if n_classes > 1 else 0)
    keep = margin.argsort(descending=True)[:n_samples]
    keep = keep.sort().values  # preserve original order
    x = x[keep].contiguous()
    labels = logits[keep].argmax(dim=1)
    y = torch.zeros(n_samples, n_classes, dtype=torch.float32)
    y[torch.arange(n_samples), labels] = 1.0
    return x, y


class Dataset:
    def __init__(self, global_batch_size: int, mubatch_size: int,
                 save_dir=None, validation: bool = False,
                 n_samples: int = 8192, in_dim: int = 784,
                 n_classes: int = 10, seed: int = 1234,
                 device=None):
        self.global_batch_size = global_batch_size
        self.mubatch_size = mubatch_size
        self.save_dir = save_dir
        self.validation = validation
        self.n_samples = n_samples
        self.in_dim = in_dim
        self.n_classes = n_classes
        self.seed = seed + (1 if validation else 0)
        self.device = torch.device(device) if device is not None else None
        self.x = None
        self.y = None

    def load(self, dp_rank: int = 0, dp_size: int = 1):
        assert self.global_batch_size % dp_size == 0, \
            "global batch must divide over DP ranks (dataset.py:35-37)"
        local_batch = self.global_batch_size // dp_size
        assert local_batch % self.mubatch_size == 0, \
            "local batch must divide into µbatches (dataset.py:60-61)"
        self.local_batch_size = local_batch

        x, y = self._load_full()
        # trim to a multiple of the global batch (dataset.py:50-52)
        n = (x.shape[0] // self.global_batch_size) * self.global_batch_size
        x, y = x[:n], y[:n]
        # strided DP shard, contiguous copy (dataset.py:54-58)
        self.x = x[dp_rank::dp_size].contiguous()
        self.y = y[dp_rank::dp_size].contiguous()
        if self.device is not None:
            self.x = self.x.to(self.device)
            self.y = self.y.to(self.device)
        # pre-cast compute copies on GPU: µbatch loads become straight
        # D2D copies instead of cast+copy (saves an elementwise kernel
        # + an allocation per µbatch on the hot path)
        if self.device is not None and self.device.type == "cuda":
            import torch as _t

            self.x_compute = self.x.to(_t.bfloat16)
            self.y_compute = self.y.to(_t.bfloat16)
        else:
            self.x_compute = self.x
            self.y_compute = self.y
        return self

    def _load_full(self):
        if self.save_dir is not None:
            split = "val" if self.validation else "train"
            xp = os.path.join(self.save_dir, f"x_{split}.parquet")
            yp = os.path.join(self.save_dir, f"y_{split}.npy")
            if os.path.exists(xp) and os.path.exists(yp):
                import pandas as pd

                x = torch.from_numpy(
                    pd.read_parquet(xp).to_numpy(dtype=np.float32))
                y = torch.from_numpy(np.load(yp).astype(np.float32))
                return x, y
        return synthesize(self.n_samples, self.in_dim, self.n_classes,
                          self.seed)

    # µbatch slicing — dataset.py:66-80
    def _mubatch_rows(self, batch_id: int, mubatch_id: int):
        start = batch_id * self.local_batch_size + mubatch_id * self.mubatch_size
        return start, start + self.mubatch_size

    def micro_batch_input(self, batch_id: int, mubatch_id: int):
        lo, hi = self._mubatch_rows(batch_id, mubatch_id)
        return self.x_compute[lo:hi]

    def micro_batch_target(self, batch_id: int, mubatch_id: int):
        lo, hi = self._mubatch_rows(batch_id, mubatch_id)
        return self.y_compute[lo:hi]

    # counts — dataset.py:82-86
    def num_batches(self) -> int:
        return self.x.shape[0] // self.local_batch_size

    def num_mubatches(self) -> int:
        return self.local_batch_size // self.mubatch_size
