"""Local-node accelerator discovery via amd-smi / rocm-smi / torch-ROCm.

MI355X-native addition (BASELINE north star: "accelerator discovery and
per-GPU telemetry come from amd-smi / the ROCm device-metrics exporter —
no NVML, no DCGM"). The reference only trusts GPU-operator node labels
(SURVEY §2.8); this module closes the loop on the node itself: it produces
the same label protocol (`amd.com/gpu.product`, `amd.com/gpu.memory`,
allocatable `amd.com/gpu`) from local ROCm tooling, so a node agent can
self-label and the emulator can mirror real hardware.

Probing order: torch.cuda (works under PyTorch-ROCm without external
binaries) → amd-smi JSON → rocm-smi. All failures degrade to None.
"""
from __future__ import annotations

import json
import re
import shutil
import subprocess
from dataclasses import dataclass
from typing import Dict, List, Optional

from ..utils.logging import get_logger

log = get_logger("discovery.local")


@dataclass
class LocalGPU:
    index: int
    name: str  # raw device marketing/arch name
    memory_mib: int
    gfx_arch: str = ""


def _product_label(gpu: LocalGPU) -> str:
    """Map a local device to the GFD product-label convention."""
    name = gpu.name.upper()
    mem_gb = round(gpu.memory_mib / 1024)
    m = re.search(r"MI\d+[A-Z]*", name)
    if m:
        return f"AMD-Instinct-{m.group(0)}-{mem_gb}GB"
    if gpu.gfx_arch.startswith("gfx950") or (270 <= mem_gb <= 300):
        return f"AMD-Instinct-MI355X-{mem_gb}GB"
    return f"AMD-{name.replace(' ', '-')}-{mem_gb}GB"


def discover_via_torch() -> Optional[List[LocalGPU]]:
    try:
        import torch

        if not torch.cuda.is_available():
            return None
        gpus = []
        for i in range(torch.cuda.device_count()):
            props = torch.cuda.get_device_properties(i)
            gpus.append(LocalGPU(
                index=i,
                name=props.name,
                memory_mib=props.total_memory // (1024 * 1024),
                gfx_arch=getattr(props, "gcnArchName", ""),
            ))
        return gpus
    except Exception as e:  # noqa: BLE001
        log.debug("torch discovery failed: %s", e)
        return None


def discover_via_amd_smi() -> Optional[List[LocalGPU]]:
    if shutil.which("amd-smi") is None:
        return None
    try:
        out = subprocess.run(
            ["amd-smi", "static", "--json"],
            capture_output=True, text=True, timeout=20, check=True,
        ).stdout
        data = json.loads(out)
        gpus = []
        entries = data if isinstance(data, list) else data.get("gpu", [])
        for i, entry in enumerate(entries):
            asic = entry.get("asic", {})
            vram = entry.get("vram", {})
            size = vram.get("size", {})
            mib = int(size.get("value", 0))
            if size.get("unit", "MB").upper().startswith("G"):
                mib *= 1024
            gpus.append(LocalGPU(
                index=i,
                name=asic.get("market_name", "AMD GPU"),
                memory_mib=mib,
                gfx_arch=asic.get("target_graphics_version", ""),
            ))
        return gpus or None
    except Exception as e:  # noqa: BLE001
        log.debug("amd-smi discovery failed: %s", e)
        return None


def discover_via_rocm_smi() -> Optional[List[LocalGPU]]:
    if shutil.which("rocm-smi") is None:
        return None
    try:
        out = subprocess.run(
            ["rocm-smi", "--showproductname", "--showmeminfo", "vram", "--json"],
            capture_output=True, text=True, timeout=20, check=True,
        ).stdout
        data = json.loads(out)
        gpus = []
        for key in sorted(k for k in data if k.startswith("card")):
            card = data[key]
            name = card.get("Card Series", card.get("Card model", "AMD GPU"))
            total = int(card.get("VRAM Total Memory (B)", 0))
            gpus.append(LocalGPU(
                index=int(key.replace("card", "") or 0),
                name=name,
                memory_mib=total // (1024 * 1024),
            ))
        return gpus or None
    except Exception as e:  # noqa: BLE001
        log.debug("rocm-smi discovery failed: %s", e)
        return None


def discover_local_gpus() -> List[LocalGPU]:
    for probe in (discover_via_torch, discover_via_amd_smi, discover_via_rocm_smi):
        gpus = probe()
        if gpus:
            return gpus
    return []


def node_labels_for_local_gpus(
    gpus: Optional[List[LocalGPU]] = None,
) -> Dict[str, str]:
    """GFD-convention labels + allocatable count for this node, suitable
    for FakeCluster Node objects or a self-labeling node agent."""
    if gpus is None:
        gpus = discover_local_gpus()
    if not gpus:
        return {}
    first = gpus[0]
    return {
        "amd.com/gpu.product": _product_label(first),
        "amd.com/gpu.memory": str(first.memory_mib),
        "amd.com/gpu.count": str(len(gpus)),
    }
