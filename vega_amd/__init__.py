"""vega_amd — MI355X-native shuffle/sort/aggregate engine (vega hot path).

Layout:
  csrc/        hand-written HIP/CDNA4 kernels + the C ABI (libvega_gpu.so)
  gpu.py       ctypes bindings (RDD-handle API + device-pointer API)
  datagen.py   deterministic synthetic input generators (host)
  shuffle.py   rank-per-GPU exchange plan (torch.distributed; RCCL on GPU)

The compute path is the HIP library only — importing this package does not
require a GPU, but using gpu.py does, and it fails loudly if the HIP
extension is missing.
"""
from . import datagen  # noqa: F401

__all__ = ["datagen", "gpu", "shuffle"]
