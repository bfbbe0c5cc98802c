"""Dataset sharding/slicing tests.  Reference: tests/test_dataset.py
(DP shard length math + dtype) — extended to µbatch arithmetic."""

import torch

from shallowspeed_amd.data import Dataset


def test_dp_shard_lengths_and_dtype():
    ds = Dataset(global_batch_size=32, mubatch_size=4, n_samples=100)
    ds.load(0, 2)
    # 100 → trim to 96 (multiple of 32) → 48 per DP rank
    assert ds.x.shape == (48, 784)
    assert ds.y.shape == (48, 10)
    assert ds.x.dtype == torch.float32
    assert ds.num_batches() == 3
    assert ds.num_mubatches() == 4


def test_shards_partition_dataset():
    a = Dataset(32, 4, n_samples=64).load(0, 2)
    b = Dataset(32, 4, n_samples=64).load(1, 2)
    full = Dataset(32, 4, n_samples=64).load(0, 1)
    # strided interleave (dataset.py:54-58)
    torch.testing.assert_close(a.x, full.x[0::2])
    torch.testing.assert_close(b.x, full.x[1::2])


def test_mubatch_slicing_arithmetic():
    ds = Dataset(16, 4, n_samples=64).load(0, 1)
    # start = batch_id*local + mubatch_id*µb (dataset.py:66-80)
    torch.testing.assert_close(ds.micro_batch_input(2, 1), ds.x[36:40])
    torch.testing.assert_close(ds.micro_batch_target(0, 3), ds.y[12:16])


def test_targets_one_hot():
    ds = Dataset(16, 4, n_samples=32).load(0, 1)
    torch.testing.assert_close(ds.y.sum(-1), torch.ones(32))


def test_file_backed_load(tmp_path):
    """Reference on-disk format: x_train.parquet + y_train.npy
    (download_dataset.py:10-23 analog written by prepare_data.py)."""
    import subprocess
    import sys

    subprocess.run([sys.executable, "prepare_data.py", "--out",
                    str(tmp_path), "--samples", "1000"], check=True)
    ds = Dataset(32, 8, save_dir=str(tmp_path), n_samples=5).load(0, 1)
    assert ds.x.shape[0] >= 800 and ds.x.shape[1] == 784  # 85% of 1000
    assert ds.y.shape[1] == 10
    val = Dataset(32, 8, save_dir=str(tmp_path), validation=True,
                  n_samples=5).load(0, 1)
    assert val.x.shape[0] <= 160
