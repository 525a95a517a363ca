"""elastic_gpu_scheduler_amd — MI355X-native Kubernetes GPU scheduler extender.

A from-scratch rebuild of the capabilities of elastic-ai/elastic-gpu-scheduler
(reference: a pure-Go kube-scheduler extender for fractional and multi-card
GPU scheduling via elasticgpu.io/gpu-core + gpu-memory extended resources),
re-designed MI355X-first:

  * per-card inventory is natively 8x MI355X / 288 GB HBM3E, sourced from
    amdsmi / rocm-smi / the in-tree HIP probe rather than NVML;
  * the allocator is xGMI-topology-aware: multi-card pods land on
    directly-linked card sets, scored by hop count;
  * the allocation hot path is native C++ (`_core`) with per-node locking —
    not a single global mutex;
  * the agent/bind side targets the ROCm k8s device plugin (amd.com/gpu) and
    the amdgpu container runtime; placement is verifiable on-device via the
    HIP `_gpuprobe` extension (gfx950).

Layer map (mirrors SURVEY.md section 1 of the rebuild blueprint):
  server/      HTTP extender protocol (filter / priorities / bind / status)
  scheduler/   resource-scheduler registry + service glue onto the C++ core
  csrc/core    native allocation core (devices, raters, topology, search)
  controller/  reconcile loop keeping the cache honest against the apiserver
  k8s/         pod/node codecs + real & fake kube clients
  agent/       MI355X node agent: inventory, topology discovery, health probe
  csrc/gpuprobe HIP gfx950 measurement kernels used by the agent
"""

from elastic_gpu_scheduler_amd.version import __version__  # noqa: F401
