"""edl_amd — MI355X-native elastic deep learning framework.

A from-scratch rebuild of the capabilities of elasticdeeplearning/edl
(reference: pure-Python control plane over PaddlePaddle Fleet/NCCL) as an
MI355X-first stack:

  * control plane: Python launcher agents + an in-repo coordination store
    (TCP, leases/watch/CAS — replaces the reference's external etcd,
    reference discovery/etcd_client.py)
  * training engine: PyTorch-ROCm one-process-per-GPU data parallel over
    RCCL/xGMI, with hand-written CDNA4 (gfx950) HIP kernels for the hot ops
  * distill plane: elastic teacher pool served by our own HIP forward path
    (reference python/edl/distill/*)

Reference layer map: SURVEY.md §1; component inventory SURVEY.md §2.
"""

__version__ = "0.1.0"
