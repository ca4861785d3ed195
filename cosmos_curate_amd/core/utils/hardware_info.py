"""GPU hardware probes — amd-smi / torch flavor.

Mirror of /root/reference/cosmos_curate/core/utils/infra/hardware_info.py
(``get_gpu_infos``, consumed by splitting_pipeline.py:181 for worker
sizing), with the NVML dependency replaced by
torch-rocm device properties plus the amd-smi CLI when present.
"""

from __future__ import annotations

import dataclasses
import shutil
import subprocess


@dataclasses.dataclass
class GpuInfo:
    index: int
    name: str
    memory_total_mb: int
    compute_units: int


def get_gpu_infos() -> list[GpuInfo]:
    """One entry per visible GPU; empty list on a GPU-less host."""
    import torch

    if not torch.cuda.is_available():
        return []
    out = []
    for i in range(torch.cuda.device_count()):
        p = torch.cuda.get_device_properties(i)
        out.append(
            GpuInfo(
                index=i,
                name=p.name,
                memory_total_mb=p.total_memory // (1024 * 1024),
                compute_units=p.multi_processor_count,
            )
        )
    return out


def amd_smi_snapshot() -> str | None:
    """Raw `amd-smi static` text for logs, None when the CLI is absent."""
    exe = shutil.which("amd-smi")
    if not exe:
        return None
    try:
        return subprocess.run(
            [exe, "static"], capture_output=True, text=True, timeout=30
        ).stdout
    except Exception:
        return None
