"""bdbnn_amd — MI355X-native binarized-NN training framework.

A from-scratch rebuild of the capabilities of the BD-BNN reference
(BlueAnon/BD-BNN: 1-bit weight/activation CNN training with a kurtosis
bimodal regularizer and teacher->student KD), designed MI355X-first:

* binary convolution as a bit-packed XNOR+popcount kernel for CDNA4
  (gfx950), with the dense stem/head on MFMA bf16 (MIOpen/rocBLAS),
* fused HIP kernels for the quantizer, kurtosis regularizer, KD losses
  and multi-tensor optimizers,
* data parallelism as one process per GPU with bucketed all-reduce over
  RCCL/xGMI overlapped with backward.

Reference parity notes cite /root/reference files as ``ref:file:line``.
"""

from . import ops
from . import models
from . import utils

__version__ = "0.1.0"
