"""Fused optimizers (K6): one multi-tensor CDNA4 HIP kernel launch per
parameter group instead of torch's per-tensor kernels.

State-dict contract: state tensors carry the SAME names,
dtypes and values as the stock torch optimizers (momentum_buffer /
exp_avg / exp_avg_sq / step), so checkpoints interchange with vanilla
PyTorch. On CPU tensors the step falls back to the identical formulas in
Python (also the numerics oracle for the GPU kernels: tests/test_gpu.py).
On a GPU box the HIP path is mandatory — a missing extension raises.
"""
from axon.optim.fused import FusedAdam, FusedAdamW, FusedLAMB, FusedSGD  # noqa: F401
