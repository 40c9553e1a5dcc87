"""GPU-share scheduler extender (binpack policy + webhook server).

The reference cooperates with an out-of-tree extender
(gpushare-scheduler-extender) purely through pod annotations.
This package ships an in-tree, wire-compatible extender so the framework is
complete on its own: the same ``ALIYUN_COM_GPU_MEM_*`` annotation handshake,
a first-fit-decreasing/best-fit binpack placement policy, and the standard
k8s scheduler-extender webhook endpoints (filter/bind).
"""

from .binpack import BinpackState, NodeGPUState
from .core import GPUShareExtender
