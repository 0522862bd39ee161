"""Translate SLURM env vars into ditask/Parallel per-process arguments.

Parity: reference ding/entry/cli_parsers/slurm_parser.py (SlurmParser:8).
"""
import os
import re
from typing import Any, Dict, Optional


class SlurmParser:

    def __init__(self, platform_spec: Optional[Dict] = None, **kwargs):
        self.kwargs = kwargs
        self.ntasks = int(os.environ['SLURM_NTASKS'])
        self.platform_spec = platform_spec
        self.ntasks_per_node = int(os.environ['SLURM_NTASKS_PER_NODE'])
        self.nodelist = self._parse_node_list()
        self.ports = int(kwargs.get('ports') or 15151)

    def _parse_node_list(self) -> list:
        raw = os.environ['SLURM_NODELIST']
        # forms: "node[01-03,05]" or "node1,node2"
        m = re.match(r'(.+?)\[(.+)\]$', raw)
        if not m:
            return raw.split(',')
        prefix, spans = m.groups()
        out = []
        for part in spans.split(','):
            if '-' in part:
                lo, hi = part.split('-')
                width = len(lo)
                out.extend(f'{prefix}{i:0{width}d}' for i in range(int(lo), int(hi) + 1))
            else:
                out.append(prefix + part)
        return out

    def parse(self) -> Dict[str, Any]:
        procid = int(os.environ['SLURM_PROCID'])
        node_rank = procid // self.ntasks_per_node
        local_rank = procid % self.ntasks_per_node
        address = self.nodelist[node_rank]
        ports = self.ports + local_rank
        attach_to = []
        if procid != 0:
            attach_to.append('tcp://{}:{}'.format(self.nodelist[0], self.ports))
        return {
            **self.kwargs,
            'address': address,
            'ports': ports,
            'node_ids': procid,
            'attach_to': attach_to,
            'labels': set(),
        }


def slurm_parser(platform_spec: Optional[Dict] = None, **kwargs) -> Dict[str, Any]:
    return SlurmParser(platform_spec, **kwargs).parse()
