"""sharedtensor_amd — an MI355X-native distributed shared-tensor engine.

A from-scratch re-engineering of the capability set of
Hello1024/shared-tensor (a Torch7/Lua + C async shared-parameter engine) for
AMD Instinct MI355X (gfx950):

  * replica + per-link residual deltas live in HBM3E
  * delta compression (1-bit sign / fp8 e4m3 / int4, all with exact
    error feedback and power-of-two scales) runs as CDNA4 HIP kernels
  * tree edges between GPUs on one node move packets over RCCL p2p (xGMI);
    TCP remains the control plane and the inter-node data plane
  * same self-organizing binary-tree topology and join-walk protocol,
    plus snapshot fast-join, reconnection, bandwidth caps, observability

Public API: SharedTensor / SharedTable / create_or_fetch (reference parity),
models.* (GPT-2 flagship workload), parallel.* (async-DP trainer).
"""
from .engine import (CODECS, SharedFlat, SharedTable, SharedTensor,
                     create_or_fetch, createOrFetch)

__version__ = "0.1.0"

__all__ = [
    "SharedTensor",
    "SharedFlat",
    "SharedTable",
    "create_or_fetch",
    "createOrFetch",
    "CODECS",
    "__version__",
]
