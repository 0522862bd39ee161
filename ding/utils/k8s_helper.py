"""Kubernetes orchestration helpers: DI-orchestrator client + launchers.

Parity: reference ding/utils/k8s_helper.py (get_operator_server_kwargs:22,
exist_operator_server:52, K8sLauncher:118) and
ding/worker/coordinator/operator_server.py (OperatorServer). The HTTP
client rides this repo's ding.interaction HttpEngine; cluster-side actions
(kubectl/k3d) shell out only when the binaries exist — the manifest and
request surfaces are fully testable offline against a stub orchestrator.
"""
import os
import shutil
import subprocess
from typing import Any, List, Optional, Tuple

from ding.utils import EasyDict

DEFAULT_NAMESPACE = 'default'
DEFAULT_POD_NAME = 'dijob-example-coordinator'
DEFAULT_API_VERSION = '/v1alpha1'


def get_operator_server_kwargs(cfg: EasyDict) -> dict:
    """Assemble OperatorServer ctor kwargs from cfg + the in-cluster env
    (KUBERNETES_POD_NAMESPACE / KUBERNETES_POD_NAME / KUBERNETES_SERVER_URL /
    KUBERNETES_SERVER_API_VERSION)."""
    cfg = cfg or EasyDict({})
    namespace = os.environ.get('KUBERNETES_POD_NAMESPACE', DEFAULT_NAMESPACE)
    name = os.environ.get('KUBERNETES_POD_NAME', DEFAULT_POD_NAME)
    url = cfg.get('system_addr', None) or os.environ.get('KUBERNETES_SERVER_URL', None)
    assert url, 'set KUBERNETES_SERVER_URL (or cfg.system_addr) on the Kubernetes platform'
    api_version = cfg.get('api_version', None) or \
        os.environ.get('KUBERNETES_SERVER_API_VERSION', DEFAULT_API_VERSION)
    if ':' in url:
        host, port = url.rsplit(':', 1)
        port = int(port)
    else:
        host, port = url, 80
    return {
        'api_version': api_version,
        'namespace': namespace,
        'name': name,
        'host': host,
        'port': port,
    }


def exist_operator_server() -> bool:
    return 'KUBERNETES_SERVER_URL' in os.environ


def pod_exec_command(kubeconfig: str, name: str, namespace: str, cmd: str) -> Tuple[int, str]:
    """kubectl exec into a pod (requires kubectl on PATH)."""
    kubectl = shutil.which('kubectl')
    if kubectl is None:
        return 1, 'kubectl not available in this image'
    proc = subprocess.run(
        [kubectl, '--kubeconfig', kubeconfig, 'exec', '-n', namespace, name, '--', 'sh', '-c', cmd],
        capture_output=True, text=True
    )
    return proc.returncode, proc.stdout + proc.stderr


class OperatorServer:
    """HTTP client for the DI-orchestrator server (replica lifecycle).

    Parity: reference ding/worker/coordinator/operator_server.py — the
    coordinator asks the orchestrator to scale collector/learner replicas
    and report failed ones.
    """

    def __init__(
        self,
        host: str,
        port: Optional[int] = None,
        api_version: str = DEFAULT_API_VERSION,
        https: bool = False,
        namespace: str = None,
        name: str = None,
    ):
        from ding.interaction.master import HttpEngine
        self._engine = HttpEngine(host, port or 80, https)
        self._api_version = api_version
        self._namespace = namespace
        self._my_name = name
        self._worker_type = None

    @property
    def api_version(self) -> str:
        return self._api_version

    def set_worker_type(self, type_: str) -> None:
        assert type_ in ('coordinator', 'aggregator'), f"invalid worker_type: {type_}"
        self._worker_type = type_

    def _path(self, path: str) -> str:
        return self._api_version + path

    @staticmethod
    def _unpack(resp: dict):
        return resp.get('code', 1) == 0, resp.get('code'), resp.get('message'), resp.get('data')

    def get_replicas(self, name: str = None):
        if name is None:
            assert self._worker_type, "set worker type first"
            params = {'namespace': self._namespace, self._worker_type: self._my_name}
        else:
            params = {'namespace': self._namespace, 'name': name}
        return self._unpack(self._engine.request('GET', self._path('/replicas'), data=params))

    def post_replicas(self, data: dict):
        data = dict(data)
        data.update({'namespace': self._namespace, 'coordinator': self._my_name})
        return self._unpack(self._engine.request('POST', self._path('/replicas'), data=data))

    def post_replicas_failed(self, collectors: List[str] = None, learners: List[str] = None):
        data = {
            'namespace': self._namespace,
            'coordinator': self._my_name,
            'collectors': collectors or [],
            'learners': learners or [],
        }
        return self._unpack(self._engine.request('POST', self._path('/replicas/failed'), data=data))

    def delete_replicas(self, n_collectors: int = 0, n_learners: int = 0):
        data = {
            'namespace': self._namespace,
            'coordinator': self._my_name,
            'collectors': {'replicas': n_collectors},
            'learners': {'replicas': n_learners},
        }
        return self._unpack(self._engine.request('DELETE', self._path('/replicas'), data=data))


ORCHESTRATOR_MANIFEST = """apiVersion: apps/v1
kind: Deployment
metadata:
  name: di-operator
  namespace: {namespace}
spec:
  replicas: 1
  selector:
    matchLabels: {{app: di-operator}}
  template:
    metadata:
      labels: {{app: di-operator}}
    spec:
      containers:
      - name: di-operator
        image: {image}
        ports:
        - containerPort: {port}
---
apiVersion: v1
kind: Service
metadata:
  name: di-server
  namespace: {namespace}
spec:
  selector: {{app: di-operator}}
  ports:
  - port: {port}
    targetPort: {port}
"""


class OrchestratorLauncher:
    """Deploy/remove the DI-orchestrator on a cluster (reference
    ding/entry/cli_ditask + orchestrator_launcher). Offline, manifest
    generation is the testable surface; create/delete shell to kubectl
    when present."""

    def __init__(
        self,
        version: str = 'v1.1.3',
        name: str = 'di-orchestrator',
        cluster: Optional[Any] = None,
        registry: str = 'opendilab',
        namespace: str = 'di-system',
        port: int = 8080,
    ):
        self.version = version
        self.name = name
        self.cluster = cluster
        self.registry = registry
        self.namespace = namespace
        self.port = port

    @property
    def image(self) -> str:
        return f'{self.registry}/di-orchestrator:{self.version}'

    def create_manifest(self, output_path: Optional[str] = None) -> str:
        manifest = ORCHESTRATOR_MANIFEST.format(namespace=self.namespace, image=self.image, port=self.port)
        if output_path:
            with open(output_path, 'w') as f:
                f.write(manifest)
        return manifest

    def _kubectl(self, *args: str) -> Tuple[int, str]:
        kubectl = shutil.which('kubectl')
        if kubectl is None:
            raise RuntimeError("kubectl unavailable offline; apply create_manifest() output on a real cluster")
        proc = subprocess.run([kubectl, *args], capture_output=True, text=True)
        return proc.returncode, proc.stdout + proc.stderr

    def create_orchestrator(self) -> None:
        import tempfile
        with tempfile.NamedTemporaryFile('w', suffix='.yaml', delete=False) as f:
            f.write(self.create_manifest())
            path = f.name
        rc, out = self._kubectl('apply', '-f', path)
        if rc != 0:
            raise RuntimeError(f'kubectl apply failed: {out}')

    def delete_orchestrator(self) -> None:
        rc, out = self._kubectl('delete', 'deployment', 'di-operator', '-n', self.namespace)
        if rc != 0:
            raise RuntimeError(f'kubectl delete failed: {out}')
