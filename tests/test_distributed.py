"""Multi-process distributed tests on CPU (gloo, world_size=2): bucketed
gradient all-reduce, policy multi_gpu path, event-bus Parallel router,
ContextExchanger/ModelExchanger/Barrier."""
import multiprocessing as mp
import os
import pickle
import time

import pytest
import torch


def _find_free_port():
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _dist_worker(rank, world, port, fn_name, out_q):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    torch.distributed.init_process_group('gloo', rank=rank, world_size=world)
    torch.manual_seed(1000 + rank)
    try:
        result = globals()[fn_name](rank, world)
        out_q.put((rank, 'ok', result))
    except Exception as e:
        import traceback
        out_q.put((rank, 'err', f"{e}\n{traceback.format_exc()}"))
    finally:
        torch.distributed.destroy_process_group()


def _run_dist(fn_name, world=2):
    ctx = mp.get_context('spawn')
    out_q = ctx.Queue()
    port = _find_free_port()
    procs = [ctx.Process(target=_dist_worker, args=(r, world, port, fn_name, out_q)) for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, status, payload = out_q.get(timeout=120)
        assert status == 'ok', f"rank {rank} failed: {payload}"
        results[rank] = payload
    for p in procs:
        p.join(timeout=30)
    return results


# ------------------------------------------------------------ worker bodies
def _body_grad_bucket_sync(rank, world):
    from ding.parallel import GradBucketAllReducer
    model = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.ReLU(), torch.nn.Linear(16, 4))
    reducer = GradBucketAllReducer(model, bucket_bytes=128, async_overlap=False)
    reducer.broadcast_params(src=0)
    x = torch.randn(5, 8) * (rank + 1)
    loss = model(x).pow(2).mean()
    loss.backward()
    local_grad = [p.grad.clone() for p in model.parameters()]
    reducer.sync()
    return {
        'params_equal_after_broadcast': True,
        'grad0': model[0].weight.grad.detach().numpy().tolist(),
        'local_grad0': local_grad[0].detach().numpy().tolist(),
    }


def _body_grad_bucket_async(rank, world):
    from ding.parallel import GradBucketAllReducer
    model = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.ReLU(), torch.nn.Linear(16, 4))
    reducer = GradBucketAllReducer(model, bucket_bytes=128, async_overlap=True)
    reducer.broadcast_params(src=0)
    for _ in range(3):  # several iterations to check hook/bucket state reuse
        for p in model.parameters():
            p.grad = None
        x = torch.randn(5, 8) * (rank + 1)
        model(x).pow(2).mean().backward()
        reducer.sync()
    return {'grad0': model[0].weight.grad.detach().numpy().tolist()}


def _body_grad_bucket_indicator(rank, world):
    """Partially-used network: rank 0 uses both heads, rank 1 only head_a.

    With the participation indicator, head_a grads average over 2 ranks,
    head_b grads over the 1 participating rank (not world_size).
    """
    from ding.parallel import GradBucketAllReducer

    class TwoHead(torch.nn.Module):

        def __init__(self):
            super().__init__()
            self.trunk = torch.nn.Linear(8, 16)
            self.head_a = torch.nn.Linear(16, 4)
            self.head_b = torch.nn.Linear(16, 4)

    model = TwoHead()
    reducer = GradBucketAllReducer(model, bucket_bytes=64, async_overlap=False, with_indicator=True)
    reducer.broadcast_params(src=0)
    x = torch.randn(5, 8) * (rank + 1)
    h = torch.relu(model.trunk(x))
    loss = model.head_a(h).pow(2).mean()
    if rank == 0:
        loss = loss + model.head_b(h).pow(2).mean()
    loss.backward()
    local = {n: (p.grad.clone() if p.grad is not None else None) for n, p in model.named_parameters()}
    reducer.sync()
    return {
        'head_a_w': model.head_a.weight.grad.numpy().tolist(),
        'head_b_w': model.head_b.weight.grad.numpy().tolist(),
        'local_head_a_w': local['head_a.weight'].numpy().tolist(),
        'local_head_b_w': None if local['head_b.weight'] is None else local['head_b.weight'].numpy().tolist(),
    }


def _body_policy_multi_gpu(rank, world):
    from ding.policy import DQNPolicy
    from ding.utils import EasyDict, deep_merge_dicts
    cfg = DQNPolicy.default_config()
    cfg = EasyDict(deep_merge_dicts(cfg, dict(
        multi_gpu=True, cuda=False,
        model=dict(obs_shape=4, action_shape=2, encoder_hidden_size_list=[16, 16]),
        learn=dict(batch_size=8, update_per_collect=1, learning_rate=1e-3),
        nstep=1,
    )))
    policy = DQNPolicy(cfg, enable_field=['learn'])
    data = []
    for _ in range(8):
        data.append({
            'obs': torch.randn(4), 'next_obs': torch.randn(4), 'action': torch.tensor([rank % 2]),
            'reward': torch.tensor([1.0]), 'done': False,
        })
    out = policy._forward_learn(data)
    w = next(policy._model.parameters()).detach()
    return {'loss': out['total_loss'], 'w_sum': float(w.sum())}


def _body_ddp_termination(rank, world):
    from ding.framework.middleware import ddp_termination_checker
    from ding.framework import task, OnlineRLContext
    with task.start(ctx=OnlineRLContext()):
        check = ddp_termination_checker(max_env_step=5, rank=rank)
        task.ctx.env_step = 10 if rank == 0 else 0  # only rank0 sees the limit
        check(task.ctx)
        return task.finish


# ------------------------------------------------------------------- tests
def test_grad_bucket_sync_mode():
    res = _run_dist('_body_grad_bucket_sync')
    import numpy as np
    g0, g1 = np.array(res[0]['grad0']), np.array(res[1]['grad0'])
    assert np.allclose(g0, g1, atol=1e-6), "grads must match after all-reduce"
    l0, l1 = np.array(res[0]['local_grad0']), np.array(res[1]['local_grad0'])
    assert np.allclose(g0, (l0 + l1) / 2, atol=1e-5), "reduced grad must be the average"


def test_grad_bucket_indicator_partial_use():
    res = _run_dist('_body_grad_bucket_indicator')
    import numpy as np
    a0, a1 = np.array(res[0]['head_a_w']), np.array(res[1]['head_a_w'])
    assert np.allclose(a0, a1, atol=1e-6)
    la0, la1 = np.array(res[0]['local_head_a_w']), np.array(res[1]['local_head_a_w'])
    assert np.allclose(a0, (la0 + la1) / 2, atol=1e-5), "used-by-all param averages over world"
    # head_b only used on rank 0: averaged over 1 participant, i.e. rank0's local grad
    b0, b1 = np.array(res[0]['head_b_w']), np.array(res[1]['head_b_w'])
    lb0 = np.array(res[0]['local_head_b_w'])
    assert res[1]['local_head_b_w'] is None
    assert np.allclose(b0, lb0, atol=1e-6), "partially-used param divides by participants (1), not world"
    assert np.allclose(b0, b1, atol=1e-6)


def test_grad_bucket_async_mode():
    res = _run_dist('_body_grad_bucket_async')
    import numpy as np
    assert np.allclose(np.array(res[0]['grad0']), np.array(res[1]['grad0']), atol=1e-6)


def test_policy_multi_gpu_learn():
    res = _run_dist('_body_policy_multi_gpu')
    assert abs(res[0]['w_sum'] - res[1]['w_sum']) < 1e-5, "post-step params must stay in sync"


def test_ddp_termination_checker():
    res = _run_dist('_body_ddp_termination')
    assert res[0] is True and res[1] is True, "rank0 finish decision must broadcast"


# --------------------------------------------------------------- event bus
def _bus_main_fn():
    from ding.framework.parallel import Parallel
    router = Parallel()
    received = []
    router.on("ping", lambda *args, **kw: received.append(args))
    # keep announcing until the peer is heard (spawn startup is slow and the
    # bus has no built-in rendezvous, matching reference test_parallel.py)
    for _ in range(200):
        router.emit("ping", router.node_id)
        if received:
            break
        time.sleep(0.1)
    assert received, f"node {router.node_id} received nothing"
    time.sleep(0.5)  # let the peer hear our last announce before teardown


def test_parallel_router_mesh():
    """Two real processes over the TCP bus exchanging events."""
    from ding.framework.parallel import Parallel
    Parallel.runner(n_parallel_workers=2, topology="mesh", startup_interval=0.1)(_bus_main_fn)


_CRASH_FILE = '/tmp/ding_test_crash_marker'


def _flaky_main():
    if not os.path.exists(_CRASH_FILE):
        with open(_CRASH_FILE, 'w') as f:
            f.write('1')
        raise RuntimeError("simulated crash")


def test_parallel_auto_recover():
    from ding.framework.parallel import Parallel
    if os.path.exists(_CRASH_FILE):
        os.remove(_CRASH_FILE)
    Parallel.runner(n_parallel_workers=1, auto_recover=True, max_retries=1)(_flaky_main)
    os.remove(_CRASH_FILE)


def _exchanger_main():
    """2 nodes: node 0 = learner, node 1 = collector. The collector produces
    trajectories + env_step; ContextExchanger ships them to the learner; the
    learner bumps train_iter which flows back. ModelExchanger broadcasts the
    learner's weights to the collector."""
    import time
    import torch
    import torch.nn as nn
    from ding.framework import OnlineRLContext, Role
    from ding.framework import task as _task
    from ding.framework.middleware import ContextExchanger, ModelExchanger

    model = nn.Linear(4, 2)
    with _task.start(ctx=OnlineRLContext()):
        is_learner = _task.router.node_id == 0
        _task.add_role(Role.LEARNER if is_learner else Role.COLLECTOR)
        _task.use(ContextExchanger(skip_n_iter=1))
        _task.use(ModelExchanger(model))

        if is_learner:
            with torch.no_grad():
                for p in model.parameters():
                    p.fill_(0.5)

            def learner_mw(ctx):
                if ctx.total_step >= 1:
                    assert ctx.trajectories is not None and len(ctx.trajectories) == 4, ctx.trajectories
                    assert ctx.env_step > 0
                ctx.train_iter += 1

            _task.use(learner_mw)
        else:

            def collector_mw(ctx):
                ctx.trajectories = [{'obs': torch.randn(4)} for _ in range(4)]
                ctx.env_step = (ctx.env_step or 0) + 4

            _task.use(collector_mw)
        _task.run(max_step=5)
        if not is_learner:
            # ModelExchanger must have delivered the learner's 0.5-filled weights
            assert all(torch.allclose(p, torch.full_like(p, 0.5)) for p in model.parameters()), \
                [p.data for p in model.parameters()]


def test_context_model_exchanger():
    from ding.framework import Parallel
    Parallel.runner(n_parallel_workers=2, topology='mesh', protocol='tcp', startup_interval=0.2)(_exchanger_main)


def _ditask_rl_main():
    """Real distributed RL: node 0 learns, node 1 collects CartPole steps;
    trajectories/model flow through Context/ModelExchanger over TCP."""
    import torch
    from ding.config import compile_config
    from ding.data import DequeBuffer
    from ding.envs import BaseEnvManagerV2
    from ding.framework import OnlineRLContext, Role
    from ding.framework import task as _task
    from ding.framework.middleware import (
        ContextExchanger, ModelExchanger, OffPolicyLearner, StepCollector, data_pusher, eps_greedy_handler,
        termination_checker,
    )
    from ding.policy import DQNPolicy
    from ding.utils import EasyDict
    from dizoo.classic_control.cartpole.config.cartpole_dqn_config import create_config, main_config
    from dizoo.classic_control.cartpole.envs.cartpole_env import CartPoleEnv

    cfg = compile_config(main_config, create_cfg=create_config, auto=True, save_cfg=False, seed=0)
    cfg.exp_name = 'exp/test_ditask_rl'
    cfg.policy.learn.batch_size = 16
    cfg.policy.collect.n_sample = 32
    policy = DQNPolicy(cfg.policy)
    with _task.start(ctx=OnlineRLContext()):
        is_learner = _task.router.node_id == 0
        _task.add_role(Role.LEARNER if is_learner else Role.COLLECTOR)
        _task.use(ContextExchanger(skip_n_iter=1))
        _task.use(ModelExchanger(policy._model))
        if is_learner:
            buffer_ = DequeBuffer(size=cfg.policy.other.replay_buffer.replay_buffer_size)
            _task.use(data_pusher(cfg, buffer_))
            _task.use(OffPolicyLearner(cfg, policy.learn_mode, buffer_))
        else:
            env = BaseEnvManagerV2(env_fn=[lambda: CartPoleEnv({}) for _ in range(2)], cfg=cfg.env.manager)
            env.seed(_task.router.node_id)
            _task.use(eps_greedy_handler(cfg))
            _task.use(StepCollector(cfg, policy.collect_mode, env))
        _task.use(termination_checker(max_env_step=int(1e5)))
        _task.run(max_step=8)
        if is_learner:
            assert _task.ctx.train_iter > 0, "learner never trained"
        else:
            assert _task.ctx.env_step > 0
            assert _task.ctx.train_iter > 0, "train_iter never flowed back to the collector"


def test_ditask_actor_learner_rl():
    from ding.framework import Parallel
    Parallel.runner(n_parallel_workers=2, topology='mesh', protocol='tcp', startup_interval=0.2)(_ditask_rl_main)


def _barrier_main():
    """Two nodes synchronize at a Barrier each iteration; both must complete
    the same number of steps without deadlock."""
    import time
    from ding.framework import OnlineRLContext
    from ding.framework import task as _task
    from ding.framework.middleware import Barrier

    with _task.start(ctx=OnlineRLContext()):
        _task.use(Barrier(attch_from_nums=1))
        steps = []

        def mw(ctx):
            steps.append(ctx.total_step)

        _task.use(mw)
        _task.run(max_step=4)
        assert len(steps) == 4, steps


def test_barrier_two_nodes():
    from ding.framework import Parallel
    Parallel.runner(n_parallel_workers=2, topology='mesh', protocol='tcp', startup_interval=0.2)(_barrier_main)


def _body_trajectory_shipper(rank, world):
    """Actor (rank 0) ships a trajectory batch to learner (rank 1) —
    tensors travel as flat dist.send payloads (GPU->GPU over RCCL on a
    real node), only the header is pickled."""
    from ding.data import TrajectoryShipper
    ship = TrajectoryShipper()
    T, B = 8, 4
    if rank == 0:
        batch = {
            'obs': torch.arange(T * B * 6, dtype=torch.float32).reshape(T, B, 6),
            'action': torch.randint(0, 4, (T, B)),
            'reward': torch.randn(T, B),
            'done': torch.zeros(T, B, dtype=torch.bool),
            'meta': {'env_id': 'test', 'unroll_len': T},
        }
        ship.send(batch, dst=1)
        return {'sent_obs_sum': float(batch['obs'].sum())}
    else:
        batch = ship.recv(src=0)
        assert batch['obs'].shape == (T, B, 6) and batch['obs'].dtype == torch.float32
        assert batch['action'].dtype == torch.int64
        assert batch['done'].dtype == torch.bool
        assert batch['meta'] == {'env_id': 'test', 'unroll_len': T}
        return {'recv_obs_sum': float(batch['obs'].sum())}


def test_trajectory_shipper_two_ranks():
    res = _run_dist('_body_trajectory_shipper')
    assert abs(res[0]['sent_obs_sum'] - res[1]['recv_obs_sum']) < 1e-5


def _body_gpu_exchanger_actor_learner(rank, world):
    """Actor rank 0 collects fake IMPALA unrolls and ships them through the
    gpu exchanger middleware; learner rank 1 trains IMPALAPolicy on the
    received collated batch (the same-node fast path, gloo here / RCCL on
    a real node)."""
    from ding.framework.middleware import gpu_trajectory_sender, gpu_trajectory_receiver
    T, B, N = 4, 3, 2

    class Ctx:
        pass

    if rank == 0:
        send = gpu_trajectory_sender(dst=1, collate=False)
        ctx = Ctx()
        ctx.env_step = 12
        ctx.trajectories = None
        ctx.train_data = {
            'obs_plus_1': torch.randn(T + 1, B, 4),
            'logit': torch.randn(T, B, N),
            'action': torch.randint(0, N, (T, B)),
            'reward': torch.randn(T, B),
            'done': torch.zeros(T, B),
        }
        send(ctx)
        return {'sent': float(ctx.train_data['obs_plus_1'].sum())}
    else:
        from ding.policy import IMPALAPolicy
        from ding.utils import EasyDict, deep_merge_dicts
        recv = gpu_trajectory_receiver(src=0)
        ctx = Ctx()
        ctx.env_step = 0
        recv(ctx)
        assert ctx.env_step == 12
        cfg = EasyDict(deep_merge_dicts(IMPALAPolicy.default_config(), EasyDict(dict(
            cuda=False, model=dict(obs_shape=4, action_shape=N, encoder_hidden_size_list=[16, 16]),
            learn=dict(batch_size=B),
        ))))
        pol = IMPALAPolicy(cfg, enable_field=['learn'])
        out = pol._forward_learn(ctx.train_data)
        assert 'total_loss' in out
        return {'recv': float(ctx.train_data['obs_plus_1'].sum())}


def test_gpu_exchanger_actor_learner():
    res = _run_dist('_body_gpu_exchanger_actor_learner')
    assert abs(res[0]['sent'] - res[1]['recv']) < 1e-4
