"""Translate DI-orchestrator/k8s pod env vars into ditask/Parallel args.

Parity: reference ding/entry/cli_parsers/k8s_parser.py.
"""
import os
from typing import Any, Dict, Optional


class K8SParser:

    def __init__(self, platform_spec: Optional[Dict] = None, **kwargs):
        self.kwargs = kwargs
        self.nodelist = os.environ.get('DI_NODES', '').split(',') if os.environ.get('DI_NODES') else []
        self.rank = int(os.environ.get('DI_RANK', os.environ.get('RANK', 0)))
        self.ports = int(kwargs.get('ports') or 15151)
        self.platform_spec = platform_spec

    def parse(self) -> Dict[str, Any]:
        address = self.nodelist[self.rank] if self.nodelist else os.environ.get('POD_IP', '127.0.0.1')
        attach_to = []
        if self.rank != 0 and self.nodelist:
            attach_to.append('tcp://{}:{}'.format(self.nodelist[0], self.ports))
        return {
            **self.kwargs,
            'address': address,
            'ports': self.ports,
            'node_ids': self.rank,
            'attach_to': attach_to,
            'labels': set(),
        }


def k8s_parser(platform_spec: Optional[Dict] = None, **kwargs) -> Dict[str, Any]:
    return K8SParser(platform_spec, **kwargs).parse()
