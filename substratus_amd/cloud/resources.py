"""Resources → Pod resources / nodeSelector mapping, MI355X-aware.

Reference internal/resources/resources.go:13-91 + gpu_info.go:14-48, extended
for MI355X: the resource name is amd.com/gpu, a single 8×MI355X node needs no
accelerator nodeSelector on kind, and admission understands 288 GB HBM3E per
GPU — a Server whose weights can't fit count × 288 GB is rejected before a
Pod is ever created (SURVEY.md §2.3 MI355X port note).
"""

from __future__ import annotations

from typing import Optional

MI355X_HBM_GB = 288
MI355X_GPUS_PER_NODE = 8


class GPUInfo:
    def __init__(self, resource_name: str, node_selector: Optional[dict] = None,
                 memory_gb: int = 0):
        self.resource_name = resource_name
        self.node_selector = node_selector or {}
        self.memory_gb = memory_gb


# cloud × GPUType → resource name + node selector
# (reference internal/resources/gpu_info.go:14-48; nvidia entries kept for
# API compatibility, amd-mi355x is the platform's first-class type)
CLOUD_GPUS: dict[tuple[str, str], GPUInfo] = {
    ("kind", "amd-mi355x"): GPUInfo("amd.com/gpu", {}, MI355X_HBM_GB),
    ("gcp", "amd-mi355x"): GPUInfo(
        "amd.com/gpu", {"cloud.google.com/gke-accelerator": "amd-mi355x"},
        MI355X_HBM_GB),
    ("aws", "amd-mi355x"): GPUInfo(
        "amd.com/gpu", {"node.kubernetes.io/instance-type": "mi355x"},
        MI355X_HBM_GB),
    ("gcp", "nvidia-l4"): GPUInfo(
        "nvidia.com/gpu", {"cloud.google.com/gke-accelerator": "nvidia-l4"}, 24),
    ("gcp", "nvidia-t4"): GPUInfo(
        "nvidia.com/gpu", {"cloud.google.com/gke-accelerator": "nvidia-tesla-t4"}, 16),
    ("gcp", "nvidia-a100"): GPUInfo(
        "nvidia.com/gpu", {"cloud.google.com/gke-accelerator": "nvidia-tesla-a100"}, 40),
    ("kind", "nvidia-l4"): GPUInfo("nvidia.com/gpu", {}, 24),
    ("kind", "nvidia-t4"): GPUInfo("nvidia.com/gpu", {}, 16),
    ("kind", "nvidia-a100"): GPUInfo("nvidia.com/gpu", {}, 40),
}


class ResourceError(ValueError):
    pass


def get_gpu_info(cloud_name: str, gpu_type: str) -> GPUInfo:
    key = (cloud_name, gpu_type)
    if key not in CLOUD_GPUS:
        raise ResourceError(f"no GPU mapping for cloud={cloud_name} "
                            f"type={gpu_type}")
    return CLOUD_GPUS[key]


def check_hbm_fit(gpu_type: str, count: int, model_bytes: int) -> None:
    """Admission: reject work whose weights cannot fit count × HBM
    (MI355X: 288 GB/GPU). model_bytes=0 skips the check."""
    if not model_bytes or count <= 0:
        return
    info = next((g for (c, t), g in CLOUD_GPUS.items() if t == gpu_type), None)
    if info is None or not info.memory_gb:
        return
    budget = info.memory_gb * count * (1 << 30)
    if model_bytes > budget * 0.95:  # leave headroom for KV cache/activations
        raise ResourceError(
            f"model needs {model_bytes / (1<<30):.0f} GiB but {count}×"
            f"{gpu_type} provides {info.memory_gb * count} GB HBM")


def apply_resources(pod_spec: dict, resources: dict, cloud_name: str) -> None:
    """Resources{cpu,memory,disk,gpu} → container requests/limits +
    nodeSelector (+ spot toleration on GCP)
    (reference internal/resources/resources.go:13-72)."""
    cpu = resources.get("cpu", 2)
    memory = resources.get("memory", 10)
    disk = resources.get("disk", 10)
    gpu = resources.get("gpu") or {}
    requests = {
        "cpu": str(cpu),
        "memory": f"{memory}Gi",
        "ephemeral-storage": f"{disk}Gi",
    }
    limits = dict(requests)
    node_selector: dict = {}
    if gpu.get("count"):
        info = get_gpu_info(cloud_name, gpu.get("type", ""))
        count = int(gpu["count"])
        if (gpu.get("type") == "amd-mi355x" and
                count > MI355X_GPUS_PER_NODE):
            raise ResourceError(
                f"gpu.count={count} exceeds the {MI355X_GPUS_PER_NODE}-GPU "
                f"xGMI node (multi-node Pods are out of scope)")
        limits[info.resource_name] = str(count)
        requests[info.resource_name] = str(count)
        node_selector.update(info.node_selector)
        if cloud_name == "gcp":
            pod_spec.setdefault("tolerations", []).append({
                "key": "cloud.google.com/gke-spot", "operator": "Equal",
                "value": "true", "effect": "NoSchedule"})
    for c in pod_spec.get("containers", []):
        c["resources"] = {"requests": requests, "limits": limits}
        if gpu.get("count"):
            # container contract "Multi-GPU": the image entrypoints run
            # `torchrun --nproc-per-node ${NUM_GPUS:-1}` (one rank per GPU
            # over RCCL) — without this env a TP=8 Server would start a
            # single rank (images/Dockerfile.model-server, -trainer)
            env = c.setdefault("env", [])
            if not any(e.get("name") == "NUM_GPUS" for e in env):
                env.append({"name": "NUM_GPUS",
                            "value": str(int(gpu["count"]))})
    if node_selector:
        pod_spec.setdefault("nodeSelector", {}).update(node_selector)


def container_builder_resources(cloud_name: str) -> dict:
    """kaniko build Pod sizing (reference resources.go:74-91): 2 CPU / 12Gi /
    100Gi ephemeral, except kind."""
    if cloud_name == "kind":
        return {}
    return {"requests": {"cpu": "2", "memory": "12Gi",
                         "ephemeral-storage": "100Gi"},
            "limits": {"ephemeral-storage": "100Gi"}}
