"""Hardware topology probe (reference include/utils/hardware_info.hpp:126,
src/utils/hardware_info.cpp — a 1,861-line CPU probe driving thread
affinity; the MI355X analog is GPU + xGMI link topology).

``HardwareInfo.probe()`` collects per-GPU properties and PARSES the
rocm-smi link topology into a peer-to-peer matrix; ``pipeline_order``
turns it into a neighbor-chained device order for pipeline-stage
placement (xGMI is point-to-point — 7 links x ~153 GB/s per MI355X — so
adjacent pipeline stages should sit on directly linked GPUs)."""

from __future__ import annotations

import os
import re
import subprocess
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

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
    # link_type[(i, j)] = "XGMI" / "PCIE" / ...; hops[(i, j)] = hop count
    link_type: Dict[Tuple[int, int], str] = field(default_factory=dict)
    hops: Dict[Tuple[int, int], int] = field(default_factory=dict)
    xgmi_links: Optional[str] = None  # raw rocm-smi output (debugging)

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
            raw = _rocm_smi("--showtopotype")
            info.xgmi_links = raw
            if raw:
                info.link_type = _parse_topo_matrix(raw)
            hops_raw = _rocm_smi("--showtopohops")
            if hops_raw:
                info.hops = {k: int(v) for k, v in
                             _parse_topo_matrix(hops_raw).items()
                             if str(v).isdigit()}
        return info

    def xgmi_peers(self, i: int) -> List[int]:
        """Direct-xGMI neighbors of GPU i."""
        return sorted(j for (a, j), t in self.link_type.items()
                      if a == i and "XGMI" in str(t).upper())

    def pipeline_order(self) -> List[int]:
        """A device order where consecutive entries are xGMI neighbors
        when the topology allows (greedy chain; identity order when no
        topology was parsed). Pipeline stage r should run on
        ``order[r]``."""
        n = len(self.gpus)
        if n <= 1 or not self.link_type:
            return list(range(n))
        unvisited = set(range(n))
        order = [0]
        unvisited.discard(0)
        while unvisited:
            cur = order[-1]
            peers = [p for p in self.xgmi_peers(cur) if p in unvisited]
            nxt = peers[0] if peers else min(unvisited)
            order.append(nxt)
            unvisited.discard(nxt)
        return order


def _rocm_smi(flag: str) -> Optional[str]:
    try:
        out = subprocess.run(["rocm-smi", flag], capture_output=True,
                             text=True, timeout=10)
        return out.stdout if out.returncode == 0 else None
    except Exception:
        return None


def _parse_topo_matrix(raw: str) -> Dict[Tuple[int, int], str]:
    """Parse rocm-smi's link matrix:

        GPU0  GPU1  ...
    GPU0 0    XGMI  ...
    GPU1 XGMI 0     ...

    Tolerates both the table form and the per-pair
    '(Topology) Link type between DRM devices i and j: XGMI' form."""
    links: Dict[Tuple[int, int], str] = {}
    header: List[int] = []
    for line in raw.splitlines():
        line = line.strip()
        m = re.match(
            r".*between DRM devices? (\d+)(?: and |.*?)(\d+)\s*:\s*(\S+)",
            line)
        if m:
            i, j, t = int(m.group(1)), int(m.group(2)), m.group(3)
            links[(i, j)] = t
            links[(j, i)] = t
            continue
        toks = line.split()
        if not toks:
            continue
        if toks[0].startswith("GPU") and all(t.startswith("GPU")
                                             for t in toks):
            header = [int(t[3:]) for t in toks]
            continue
        if toks[0].startswith("GPU") and header and len(toks) == len(header) + 1:
            try:
                i = int(toks[0][3:])
            except ValueError:
                continue
            for j, val in zip(header, toks[1:]):
                if i != j:
                    links[(i, j)] = val
    return links
