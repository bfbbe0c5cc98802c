"""shallowspeed_amd — an MI355X-native minimal distributed-training engine.

A from-scratch rebuild of the capabilities of siboehm/ShallowSpeed
(reference: /root/reference, a ~1.6 kLoC NumPy+mpi4py proof-of-concept)
as an AMD MI355X (gfx950, CDNA4) native framework:

  * framework layer: PyTorch-ROCm tensors and autograd-free explicit
    forward/backward modules (reference: shallowspeed/layers.py)
  * compute layer: hand-written HIP/CDNA4 kernels (MFMA bf16 GEMMs with
    fused bias+ReLU epilogues, split-K atomic wgrad with fused bias-grad,
    fused softmax-cross-entropy / softmax-MSE loss heads, multi-tensor
    SGD emitting bf16 + transposed-bf16 weight copies)
    (reference compute: shallowspeed/functional.py -> NumPy BLAS)
  * communication layer: RCCL over xGMI via torch.distributed
    ("nccl" backend == RCCL on ROCm), bucketed gradient all-reduce
    overlapped with backward, grouped p2p for pipeline edges
    (reference comms: mpi4py, shallowspeed/pipe.py:302-327, 367-381)
  * orchestration: schedule-as-instruction-stream design — Schedule
    objects emit pure-data instruction lists which a Worker interprets
    (reference: shallowspeed/pipe.py:141-466), including a real
    PipeDream-Flush / 1F1B schedule (stubbed in the reference at
    pipe.py:297-299).
"""

__version__ = "0.1.0"

from . import utils  # noqa: F401
