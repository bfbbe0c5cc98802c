"""Rank-0 printing, model hashing, replica-sync assertion.

Reference: shallowspeed/utils.py:8-31 (rprint / get_model_hash /
assert_sync over mpi4py) — rebuilt on torch.distributed.
"""

import hashlib

import torch


def rprint(*args, **kwargs):
    """Print only on (global) rank 0.  Reference: utils.py:8-10."""
    import torch.distributed as dist

    if not dist.is_available() or not dist.is_initialized() or dist.get_rank() == 0:
        print(*args, **kwargs)


def get_model_hash(model) -> str:
    """sha1 over every parameter's f32 master bytes, concatenated, then
    re-hashed.  Reference: utils.py:13-24."""
    hashes = []
    for p in model.parameters():
        b = p.data.detach().float().cpu().contiguous().numpy().tobytes()
        hashes.append(hashlib.sha1(b).hexdigest())
    return hashlib.sha1("".join(hashes).encode()).hexdigest()


def assert_sync(group, model_hash: str):
    """Gather replica hashes in the DP group and assert equality.
    Reference: utils.py:27-31 + train.py:154-155."""
    import torch.distributed as dist

    if group is None or not dist.is_initialized():
        return
    world = dist.get_world_size(group=group)
    if world == 1:
        return
    out = [None] * world
    dist.all_gather_object(out, model_hash, group=group)
    if len(set(out)) != 1:
        raise RuntimeError(f"DP replicas out of sync: {out}")


class StepTimer:
    """Wall-clock timer with optional device sync (the reference only
    had per-epoch prints, train.py:131-137; this is the per-instruction
    timing hook surface)."""

    def __init__(self, device=None):
        self.device = device
        self.reset()

    def reset(self):
        self._t0 = None
        self.elapsed = 0.0

    def __enter__(self):
        import time

        if self.device is not None and torch.device(self.device).type == "cuda":
            torch.cuda.synchronize(self.device)
        self._t0 = time.perf_counter()
        return self

    def __exit__(self, *exc):
        import time

        if self.device is not None and torch.device(self.device).type == "cuda":
            torch.cuda.synchronize(self.device)
        self.elapsed += time.perf_counter() - self._t0
        return False
