"""MI355X-native dynamic-batch-size data-parallel trainer ("DBS").

A from-scratch rebuild of the capabilities of
Soptq/Dynamic_Load_Balance_DistributedDNN (the official implementation of
"DBS: Dynamic Batch Size For Distributed Deep Neural Network Training",
Ye et al. 2020) designed MI355X-first:

- one process per GPU, `torch.distributed` over RCCL (xGMI) for all
  collectives; gloo only for the CPU debug path (`-d true`);
- the DBS load balancer times each worker's pure compute with hipEvents and
  re-partitions the global batch with an *exact-sum* integer allocator
  (the reference's float-truncation scheme at dataloader.py:43-45 /
  dbs.py:465-473 can deadlock ranks with skewed iteration counts);
- gradient averaging is a batch-share-weighted all-reduce, bucketed on a
  flat gradient arena and overlapped with backward;
- the model zoo's hot ops (conv / GroupNorm+ReLU / attention / LayerNorm /
  softmax losses) are hand-written CDNA4 HIP kernels (MFMA + LDS tiling)
  compiled for gfx950.

Public surface mirrors the reference's `dbs.py` CLI (13 flags, same
defaults) and artifact layout (`./logs/*.log`, `./statis/*.npy`).
"""

__version__ = "0.1.0"

from . import scheduler  # noqa: F401
