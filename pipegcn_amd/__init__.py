"""pipegcn_amd — an MI355X-native full-graph GCN training engine.

A from-scratch reimplementation of the capabilities of GATECH-EIC/PipeGCN
(pipelined full-graph GNN training with stale boundary feature/gradient
exchange), designed MI355X-first:

 - hand-written HIP/CDNA4 (gfx950) kernels for the hot path (CSR SpMM with
   fused degree-divide, row gather/scatter, EMA correction) — no DGL, no CUDA,
   no Triton, no dual path;
 - RCCL over xGMI (torch.distributed backend "nccl" on ROCm) for GPU-direct
   boundary exchange on a side HIP stream and for gradient/BN collectives,
   with a gloo/CPU path behind the same transport interface for GPU-less
   plumbing tests;
 - our own C++ graph core (CSR construction, partitioner) replacing
   DGL/METIS.

CLI and checkpoint format follow the reference exactly
(/root/reference/helper/parser.py, /root/reference/train.py:397).
"""

__version__ = "0.1.0"

import torch  # noqa: F401  (must load libtorch before the extension)

try:
    from pipegcn_amd import _C  # noqa: F401

    HAS_NATIVE = True
except ImportError:  # pragma: no cover - build environments only
    _C = None
    HAS_NATIVE = False


def native():
    """Return the native extension, raising loudly if it was not built."""
    if _C is None:
        raise ImportError(
            "pipegcn_amd._C native extension is not built. "
            "Run `python setup.py build_ext --inplace` "
            "(or `python -c 'import __graft_entry__; __graft_entry__.build()'`)."
        )
    return _C
