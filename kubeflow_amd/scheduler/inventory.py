"""GPU/NUMA inventory for one 8×MI355X node.

Replaces kubelet's node-status/device-plugin view. Sources, in order:
  * KF_FAKE_GPUS env (CPU CI: pretend N GPUs so scheduler tests run here),
  * torch.cuda (ROCm) device enumeration + properties,
  * amd-smi/rocm-smi fall-back parsing for NUMA affinity.
"""
from __future__ import annotations

import os
import subprocess
from dataclasses import dataclass
from typing import List


@dataclass
class GpuInfo:
    index: int
    name: str = "MI355X"
    hbm_bytes: int = 288 * 1024**3
    numa_node: int = 0


class GpuInventory:
    def __init__(self):
        self.gpus: List[GpuInfo] = []
        fake = os.environ.get("KF_FAKE_GPUS")
        if fake is not None:
            self.gpus = [GpuInfo(index=i) for i in range(int(fake))]
            self.is_fake = True
            return
        self.is_fake = False
        try:
            import torch
            if torch.cuda.is_available():
                for i in range(torch.cuda.device_count()):
                    props = torch.cuda.get_device_properties(i)
                    self.gpus.append(GpuInfo(
                        index=i, name=props.name,
                        hbm_bytes=props.total_memory,
                        numa_node=self._numa_for(i)))
        except Exception:
            pass

    @staticmethod
    def _numa_for(index: int) -> int:
        """NUMA node of a GPU via sysfs (best-effort; 0 if unknown)."""
        try:
            out = subprocess.run(
                ["rocm-smi", "--showtoponuma", "--json"],
                capture_output=True, text=True, timeout=5)
            import json
            data = json.loads(out.stdout)
            card = data.get(f"card{index}", {})
            for k, v in card.items():
                if "numa" in k.lower() and "node" in k.lower():
                    return int(v)
        except Exception:
            pass
        return 0

    @property
    def n_gpus(self) -> int:
        return len(self.gpus)

    def describe(self) -> List[dict]:
        return [{"index": g.index, "name": g.name, "hbm_bytes": g.hbm_bytes,
                 "numa_node": g.numa_node} for g in self.gpus]
