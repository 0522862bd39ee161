"""Aux utils: autolog, loader DSL, profiler, memory profiler, dataloader,
scheduler, normalizer, k8s/slurm generation."""
import numpy as np
import pytest
import torch

from ding.utils import (
    LoggedModel, LoggedValue, TickTime, Loader, LoaderError, is_type, interval, enum, dict_, collection, optional,
    Profiler, SimpleMemoryProfiler, Scheduler, DatasetNormalizer, K8sLauncher, generate_slurm_script, EasyDict,
)


def test_autolog():
    class Mon(LoggedModel):
        v = LoggedValue(float)

        def __init__(self, t, expire):
            super().__init__(t, expire)

    t = TickTime()
    m = Mon(t, expire=5)
    for i in range(10):
        m.v = float(i)
        t.step()
    vals = [v for _, v in m.range_values('v')]
    assert len(vals) <= 6  # window pruned
    assert m.avg('v') > 0 and m.max('v') == 9.0


def test_loader_dsl():
    pos_int = is_type(int) & interval(0, None)
    assert pos_int(5) == 5
    with pytest.raises(Exception):
        pos_int(-1)
    with pytest.raises(Exception):
        pos_int("x")
    color = enum('red', 'green')
    assert color('red') == 'red'
    schema = dict_(lr=is_type(float) & interval(0, 1), n=is_type(int))
    out = schema({'lr': 0.1, 'n': 3, 'extra': 1})
    assert out == {'lr': 0.1, 'n': 3}
    assert collection(is_type(int))([1, 2]) == [1, 2]
    assert optional(is_type(int))(None) is None
    piped = is_type(int) | is_type(float)
    assert piped(0.5) == 0.5


def test_profilers(tmp_path):
    p = Profiler()
    p.mkdir(str(tmp_path / 'prof'))
    net = torch.nn.Sequential(torch.nn.Linear(4, 8), torch.nn.ReLU(), torch.nn.Linear(8, 2))
    mp = SimpleMemoryProfiler(net, log_folder=str(tmp_path / 'mem'), total_steps=1)
    net(torch.randn(3, 4))
    mp.step()
    assert (tmp_path / 'mem' / 'params.txt').exists()
    assert (tmp_path / 'mem' / 'activations.txt').exists()


def test_async_dataloader():
    from ding.utils.data import AsyncDataLoader
    def source(bs):
        return [{'x': torch.randn(3), 'y': 1} for _ in range(bs)]
    loader = AsyncDataLoader(source, batch_size=4, num_workers=1)
    batch = next(loader)
    assert batch['x'].shape == (4, 3)
    loader.close()


def test_scheduler_and_normalizer():
    s = Scheduler(EasyDict(dict(schedule_mode='reduce', factor=0.1, change_range=[0, 1], patience=1)))
    param = 0.5
    for _ in range(5):
        param = s.step(10.0, param)  # flat metric -> reduce after patience
    assert param < 0.5
    data = {'obs': np.random.randn(100, 4).astype(np.float32)}
    norm = DatasetNormalizer(data, 'gaussian')
    z = norm.normalize(data['obs'], 'obs')
    assert abs(z.mean()) < 0.1
    back = norm.unnormalize(z, 'obs')
    assert np.allclose(back, data['obs'], atol=1e-4)


def test_cluster_script_generation(tmp_path):
    k8s = K8sLauncher()
    manifest = k8s.create_manifest('test-job', 'pkg.main', workers=2, output_path=str(tmp_path / 'job.yaml'))
    assert 'parallelism: 2' in manifest
    script = generate_slurm_script('job', 'python bench.py', nodes=2, output_path=str(tmp_path / 'job.sh'))
    assert '#SBATCH --nodes=2' in script


def test_gpu_prioritized_buffer_cpu_semantics():
    """GPUPrioritizedBuffer math runs on CPU device too (device='cpu'):
    ring wrap, prioritized sampling bias, IS weights, priority updates."""
    import torch
    from ding.data import GPUPrioritizedBuffer
    buf = GPUPrioritizedBuffer(size=64, alpha=1.0, beta=1.0, device='cpu')
    batch = {'obs': torch.randn(32, 4), 'reward': torch.randn(32)}
    idx = buf.push(batch)
    assert buf.count() == 32 and idx.shape == (32, )
    # second push wraps the ring
    buf.push({'obs': torch.randn(48, 4), 'reward': torch.randn(48)})
    assert buf.count() == 64
    # bias: give one index huge priority, it should dominate samples
    hot = torch.tensor([5])
    buf.update_priority(hot, torch.tensor([1000.0]))
    sampled, sidx, w = buf.sample(256)
    assert sampled['obs'].shape == (256, 4)
    frac_hot = (sidx == 5).float().mean()
    assert frac_hot > 0.5, float(frac_hot)
    assert w.max() <= 1.0 + 1e-6
    # hot samples get the smallest IS weight
    assert torch.isclose(w[sidx == 5].max(), w.min())
    sd = buf.state_dict()
    buf2 = GPUPrioritizedBuffer(size=64, device='cpu')
    buf2.load_state_dict(sd)
    assert buf2.count() == 64


@pytest.mark.benchmark
def test_buffer_throughput_benchmark():
    """Buffer micro-benchmark (SURVEY §4 benchmark-marked buffer test):
    push/sample throughput of DequeBuffer; loose sanity floor only."""
    import time
    import torch
    from ding.data import DequeBuffer
    buf = DequeBuffer(size=10000)
    data = [{'obs': torch.randn(8), 'action': 1, 'reward': 0.5} for _ in range(1000)]
    t0 = time.time()
    for d in data * 5:
        buf.push(d)
    push_rate = 5000 / (time.time() - t0)
    t0 = time.time()
    for _ in range(50):
        buf.sample(256)
    sample_rate = 50 * 256 / (time.time() - t0)
    assert push_rate > 1e3 and sample_rate > 1e3, (push_rate, sample_rate)


def test_wandb_logger_real_path_with_stub():
    """wandb loggers use the real wandb API when importable (stubbed here)."""
    import sys, types
    calls = []
    stub = types.ModuleType('wandb')
    stub.run = None
    stub.init = lambda **kw: (calls.append(('init', kw)), setattr(stub, 'run', object()))
    stub.log = lambda payload, step=None: calls.append(('log', payload, step))
    stub.watch = lambda m: calls.append(('watch', ))
    old = sys.modules.get('wandb')
    sys.modules['wandb'] = stub
    try:
        import importlib
        import ding.framework.middleware.functional.logger as L
        from ding.framework import OnlineRLContext
        lg = L.wandb_online_logger(project_name='t', run_name='r')
        ctx = OnlineRLContext()
        ctx.env_step, ctx.train_iter = 10, 2
        ctx.train_output = {'total_loss': 1.5}
        ctx.eval_value = 3.0
        lg(ctx)
    finally:
        if old is None:
            sys.modules.pop('wandb', None)
        else:
            sys.modules['wandb'] = old
    payload = [c for c in calls if c[0] == 'log'][0][1]
    assert payload['train/total_loss'] == 1.5
    assert payload['eval/episode_return'] == 3.0


def test_wandb_logger_fallback_without_wandb():
    import ding.framework.middleware.functional.logger as L
    from ding.framework import OnlineRLContext
    lg = L.wandb_online_logger()  # wandb absent in this image -> JSONL lane
    ctx = OnlineRLContext()
    ctx.env_step, ctx.train_iter = 1, 1
    ctx.train_output = {'total_loss': 0.1}
    ctx.eval_value = float('-inf')
    lg(ctx)  # must not raise


def test_operator_server_against_stub_orchestrator():
    """OperatorServer client round-trips replicas API against a local stub
    DI-orchestrator (flask)."""
    import threading
    import time
    from flask import Flask, jsonify, request as freq
    from ding.utils import OperatorServer
    from ding.utils.misc_helpers import find_free_port

    app = Flask('stub_orchestrator')
    state = {'collectors': [], 'learners': [], 'failed': []}

    @app.route('/v1alpha1/replicas', methods=['GET', 'POST', 'DELETE'])
    def replicas():
        data = freq.get_json(silent=True) or {}
        if freq.method == 'POST':
            state['collectors'] = [f"cl-{i}" for i in range(int(data.get('collectors', 0)))]
            state['learners'] = [f"ln-{i}" for i in range(int(data.get('learners', 0)))]
        if freq.method == 'DELETE':
            state['collectors'] = state['collectors'][:-int(data['collectors']['replicas']) or None]
        return jsonify({'code': 0, 'message': 'success',
                        'data': {'collectors': state['collectors'], 'learners': state['learners']}})

    @app.route('/v1alpha1/replicas/failed', methods=['POST'])
    def failed():
        data = freq.get_json(silent=True) or {}
        state['failed'] = data.get('collectors', []) + data.get('learners', [])
        return jsonify({'code': 0, 'message': 'success', 'data': state['failed']})

    port = find_free_port()
    th = threading.Thread(target=lambda: app.run(host='127.0.0.1', port=port), daemon=True)
    th.start()
    time.sleep(0.5)

    srv = OperatorServer(host='127.0.0.1', port=port, namespace='di-test', name='coord-0')
    srv.set_worker_type('coordinator')
    ok, code, msg, data = srv.post_replicas({'collectors': 2, 'learners': 1})
    assert ok and len(data['collectors']) == 2 and len(data['learners']) == 1
    ok, _, _, data = srv.get_replicas()
    assert ok and data['collectors'] == ['cl-0', 'cl-1']
    ok, _, _, data = srv.post_replicas_failed(collectors=['cl-1'])
    assert ok and data == ['cl-1']
    ok, _, _, data = srv.delete_replicas(n_collectors=1)
    assert ok and len(data['collectors']) == 1


def test_k8s_env_kwargs_and_manifests(tmp_path, monkeypatch):
    from ding.utils import get_operator_server_kwargs, exist_operator_server, OrchestratorLauncher
    from ding.utils import EasyDict
    monkeypatch.setenv('KUBERNETES_SERVER_URL', 'di-server.di-system:8080')
    monkeypatch.setenv('KUBERNETES_POD_NAMESPACE', 'ns1')
    monkeypatch.setenv('KUBERNETES_POD_NAME', 'job-coordinator')
    assert exist_operator_server()
    kw = get_operator_server_kwargs(EasyDict({}))
    assert kw == {'api_version': '/v1alpha1', 'namespace': 'ns1', 'name': 'job-coordinator',
                  'host': 'di-server.di-system', 'port': 8080}
    launcher = OrchestratorLauncher(namespace='di-system')
    manifest = launcher.create_manifest(output_path=str(tmp_path / 'orch.yaml'))
    assert 'di-operator' in manifest and 'di-server' in manifest
    assert (tmp_path / 'orch.yaml').exists()


def test_dist_entry_replica_commands_against_stub():
    """dist_{add,delete,restart}_replicas drive the orchestrator API."""
    import threading
    import time
    from flask import Flask, jsonify, request as freq
    from ding.entry import dist_add_replicas, dist_delete_replicas, dist_restart_replicas
    from ding.utils import EasyDict
    from ding.utils.misc_helpers import find_free_port

    app = Flask('stub_orch2')
    state = {'collectors': 0, 'failed': []}

    @app.route('/v1alpha1/replicas', methods=['POST', 'DELETE'])
    def replicas():
        data = freq.get_json(silent=True) or {}
        if freq.method == 'POST':
            state['collectors'] += int(data.get('collectors', 0))
        else:
            state['collectors'] -= int(data['collectors']['replicas'])
        return jsonify({'code': 0, 'message': 'success', 'data': {'collectors': state['collectors']}})

    @app.route('/v1alpha1/replicas/failed', methods=['POST'])
    def failed():
        data = freq.get_json(silent=True) or {}
        state['failed'] = data.get('collectors', [])
        return jsonify({'code': 0, 'message': 'success', 'data': state['failed']})

    port = find_free_port()
    threading.Thread(target=lambda: app.run(host='127.0.0.1', port=port), daemon=True).start()
    time.sleep(0.5)
    cfg = EasyDict(dict(system=dict(system_addr=f'127.0.0.1:{port}')))
    assert dist_add_replicas(cfg, n_collectors=3)['collectors'] == 3
    assert dist_delete_replicas(cfg, n_collectors=1)['collectors'] == 2
    assert dist_restart_replicas(cfg, collectors=['cl-0']) == ['cl-0']
