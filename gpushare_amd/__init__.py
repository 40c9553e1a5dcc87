"""gpushare_amd — MI355X-native Kubernetes GPU-sharing device plugin.

A from-scratch rebuild of the capabilities of
AliyunContainerService/gpushare-device-plugin (reference: Go + NVML/CUDA
ecosystem) as an AMD-native framework for MI355X (gfx950) nodes:

- GPU enumeration and per-device HBM3E capacity (288 GiB / GPU) come from
  an in-tree dlopen shim over ROCm's libamd_smi.so plus the KFD sysfs
  topology (/sys/class/kfd/kfd/topology) — no NVML, no CUDA shims.
- Each GiB (or MiB) of GPU memory is advertised as one schedulable fake
  device under the extended resource ``aliyun.com/gpu-mem`` over the
  kubelet device-plugin gRPC API v1beta1 (wire-compatible with the
  reference and with the gpushare-scheduler-extender annotation protocol).
- ``Allocate`` resolves the scheduler-extender's pod-annotation binding to
  a physical GPU and injects ``/dev/kfd`` + the GPU's ``/dev/dri/renderD*``
  device nodes and ``HIP_VISIBLE_DEVICES`` / ``ROCR_VISIBLE_DEVICES`` so
  co-located PyTorch-ROCm workloads share one CDNA4 GPU.

Reference layer map: see SURVEY.md at the repo root.
"""

__version__ = "0.1.0"
