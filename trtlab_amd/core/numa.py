"""NUMA topology walk (reference trtlab/core/src/affinity.cc:1-270:
cpuaff-based cpu_set algebra + /sys distances; SURVEY.md §2.2 item
"affinity / NUMA"). Parses /sys/devices/system/node: per-node cpu lists,
inter-node distance matrix, and memory size — the inputs for pinning
worker threads and first-touch allocation near the right GPU's host
bridge."""
from __future__ import annotations

import os
from dataclasses import dataclass, field
from typing import Dict, List

_SYS = "/sys/devices/system/node"


def _parse_cpulist(s: str) -> List[int]:
    cpus: List[int] = []
    for part in s.strip().split(","):
        if not part:
            continue
        if "-" in part:
            a, b = part.split("-")
            cpus.extend(range(int(a), int(b) + 1))
        else:
            cpus.append(int(part))
    return cpus


@dataclass
class NumaNode:
    id: int
    cpus: List[int] = field(default_factory=list)
    distances: List[int] = field(default_factory=list)  # to node 0..N-1
    mem_total_kb: int = 0


class NumaTopology:
    """system topology (reference system::topology, affinity.h:107)."""

    def __init__(self, sysfs: str = _SYS):
        self.nodes: Dict[int, NumaNode] = {}
        if not os.path.isdir(sysfs):
            return  # non-NUMA / restricted container: empty topology
        for entry in sorted(os.listdir(sysfs)):
            if not entry.startswith("node") or not entry[4:].isdigit():
                continue
            nid = int(entry[4:])
            node = NumaNode(nid)
            base = os.path.join(sysfs, entry)
            try:
                with open(os.path.join(base, "cpulist")) as f:
                    node.cpus = _parse_cpulist(f.read())
            except OSError:
                pass
            try:
                with open(os.path.join(base, "distance")) as f:
                    node.distances = [int(v) for v in f.read().split()]
            except OSError:
                pass
            try:
                with open(os.path.join(base, "meminfo")) as f:
                    for line in f:
                        if "MemTotal" in line:
                            node.mem_total_kb = int(line.split()[-2])
                            break
            except OSError:
                pass
            self.nodes[nid] = node

    def node_of_cpu(self, cpu: int) -> int:
        for nid, n in self.nodes.items():
            if cpu in n.cpus:
                return nid
        return 0

    def nearest_cpus(self, node_id: int) -> List[int]:
        """CPUs of `node_id`, then of other nodes by increasing distance
        (the pin-order for worker pools serving a device on that node)."""
        if node_id not in self.nodes:
            return sorted(c for n in self.nodes.values() for c in n.cpus)
        me = self.nodes[node_id]
        order = sorted(
            self.nodes.values(),
            key=lambda n: me.distances[n.id]
            if n.id < len(me.distances) else 255)
        out: List[int] = []
        for n in order:
            out.extend(n.cpus)
        return out
