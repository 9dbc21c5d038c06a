"""Hardware topology probe (reference include/utils/hardware_info.hpp:126).

The reference probes CPU topology to drive thread affinity; the MI355X
analog is GPU + xGMI link topology from torch/HIP device properties and
rocm-smi, used for pipeline-stage placement sanity checks.
"""

from __future__ import annotations

import os
import subprocess
from dataclasses import dataclass, field
from typing import List, Optional

import torch


@dataclass
class GPUInfo:
    index: int
    name: str
    total_memory: int
    multi_processor_count: int
    gcn_arch: str = ""


@dataclass
class HardwareInfo:
    cpu_count: int = 0
    gpus: List[GPUInfo] = field(default_factory=list)
    xgmi_links: Optional[str] = None

    @classmethod
    def probe(cls) -> "HardwareInfo":
        info = cls(cpu_count=os.cpu_count() or 1)
        if torch.cuda.is_available():
            for i in range(torch.cuda.device_count()):
                p = torch.cuda.get_device_properties(i)
                info.gpus.append(GPUInfo(
                    index=i, name=p.name, total_memory=p.total_memory,
                    multi_processor_count=p.multi_processor_count,
                    gcn_arch=getattr(p, "gcnArchName", "")))
            info.xgmi_links = _rocm_smi_topology()
        return info


def _rocm_smi_topology() -> Optional[str]:
    try:
        out = subprocess.run(["rocm-smi", "--showtopotype"],
                             capture_output=True, text=True, timeout=10)
        return out.stdout if out.returncode == 0 else None
    except Exception:
        return None
