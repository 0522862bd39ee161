"""Flagship benchmark: Atari PPO learner throughput on MI355X.

Measures the BASELINE.json headline metric — learner samples/sec
(env-steps/sec) for the Atari Pong PPO workload (reference config
dizoo/atari/config/serial/pong/pong_ppo_config.py: n_sample=3200,
batch=320, epoch_per_collect=10, conv encoder [64,64,128] on 4x84x84
frames) — on synthetic env transitions and random-init weights (no network
for datasets). One bench step = one full PPO train phase over a freshly
generated 3200-sample batch: 10 epochs x 10 minibatches of 320, with
advantage recomputation each epoch, forward+backward+optimizer step all
inside the timed region. fp32 compute (same precision as the reference's
training).

Also supports the IMPALA workload (--workload impala): unroll_len=32,
batch=128 trajectories, v-trace loss (reference
spaceinvaders_impala_config.py:17-45).

Multi-GPU: launched by the driver via torch.distributed.run with one rank
per GPU over RCCL; weak scaling (each rank trains its own 3200-sample
batch; gradients all-reduced through the bucketed reducer).
"""
import argparse
import json
import os
import time

import os
# MIOpen find-mode: force tuned-solver search+db persist (search runs during
# untimed warmup; A/B on MI355X: 13.5k vs 12.6k samples/s on the PPO bench)
os.environ.setdefault('MIOPEN_FIND_ENFORCE', '3')
import torch

# MIOpen autotune: pick the fastest conv algorithms for the fixed bench shapes
torch.backends.cudnn.benchmark = True


def _channels_last_on() -> bool:
    # NHWC default: MIOpen's channels-last solvers measured 181.8 vs 211.5
    # ms/step on the PPO bench (A/B bench_nhwc.json / bench_nchw.json)
    return os.environ.get('DING_CHANNELS_LAST', '1') not in ('0', 'false')


def _maybe_channels_last(policy):
    if _channels_last_on():
        policy._model.to(memory_format=torch.channels_last)
    return policy


def build_ppo_policy(device: str, multi_gpu: bool, bf16: bool = False):
    from ding.policy import PPOPolicy
    from ding.utils import EasyDict, deep_merge_dicts

    cfg = PPOPolicy.default_config()
    user = dict(
        cuda=device.startswith("cuda"),
        multi_gpu=multi_gpu,
        action_space='discrete',
        recompute_adv=True,
        model=dict(
            obs_shape=[4, 84, 84],
            action_shape=6,
            action_space='discrete',
            encoder_hidden_size_list=[64, 64, 128],
            actor_head_hidden_size=128,
            critic_head_hidden_size=128,
        ),
        learn=dict(
            epoch_per_collect=10,
            batch_size=320,
            learning_rate=3e-4,
            value_weight=0.5,
            entropy_weight=0.001,
            clip_ratio=0.2,
            adv_norm=True,
            value_norm=True,
            ignore_done=False,
            grad_clip_type='clip_norm',
            grad_clip_value=0.5,
            bf16=bf16,
            # hipGraph-capture the minibatch fwd+loss+bwd (launch-bound step;
            # single-process only — the policy ignores this under multi_gpu;
            # under bf16 the autocast weight-casts are captured into the graph)
            cuda_graph=os.environ.get('DING_PPO_GRAPH', '1') not in ('0', 'false'),
            channels_last=_channels_last_on(),
        ),
        collect=dict(n_sample=3200, unroll_len=1, discount_factor=0.99, gae_lambda=0.95),
    )
    cfg = EasyDict(deep_merge_dicts(cfg, user))
    return _maybe_channels_last(PPOPolicy(cfg, enable_field=['learn']))


def make_ppo_batch(n_sample: int, device: str, policy) -> dict:
    """Synthetic Pong-shaped transitions resident on the GPU."""
    obs = torch.randint(0, 255, (n_sample, 4, 84, 84), dtype=torch.uint8, device=device)
    next_obs = torch.randint(0, 255, (n_sample, 4, 84, 84), dtype=torch.uint8, device=device)
    with torch.no_grad():
        # behaviour logits/values from the current net (realistic distributions)
        out = policy._model(obs[:320].float().div_(255.0), mode='compute_actor_critic')
    B = n_sample
    logit = out['logit'].detach().repeat((B + 319) // 320, 1)[:B].contiguous()
    value = out['value'].detach().repeat((B + 319) // 320)[:B].contiguous()
    # gumbel-max sample: sync-free (Categorical's arg validation forces a
    # device->host sync per step)
    gumbel = -torch.log(-torch.log(torch.rand_like(logit) + 1e-10) + 1e-10)
    action = (logit + gumbel).argmax(dim=-1)
    reward = torch.sign(torch.randn(B, device=device)) * (torch.rand(B, device=device) < 0.05)
    done = (torch.rand(B, device=device) < 0.002).float()
    return {
        'obs': obs,
        'next_obs': next_obs,
        'action': action,
        'logit': logit,
        'value': value,
        'adv': torch.randn(B, device=device),
        'reward': reward,
        'done': done,
        'weight': None,
    }


def ppo_step(policy, device: str, n_sample: int):
    data = make_ppo_batch(n_sample, device, policy)
    # scale obs like the Atari pipeline (uint8 -> [0,1]) without a CPU trip
    data['obs'] = data['obs'].float().div_(255.0)
    data['next_obs'] = data['next_obs'].float().div_(255.0)
    if _channels_last_on():
        data['obs'] = data['obs'].to(memory_format=torch.channels_last)
        data['next_obs'] = data['next_obs'].to(memory_format=torch.channels_last)
    policy._forward_learn(data)
    return n_sample


def build_impala(device: str, multi_gpu: bool, bf16: bool = False):
    """IMPALA workload through the real IMPALAPolicy (reference
    spaceinvaders_impala_config.py:17-45: unroll_len=32, batch=128,
    encoder [128,128,256])."""
    from ding.policy import IMPALAPolicy
    from ding.utils import EasyDict, deep_merge_dicts

    cfg = IMPALAPolicy.default_config()
    user = dict(
        cuda=device.startswith('cuda'),
        multi_gpu=multi_gpu,
        unroll_len=32,
        model=dict(
            obs_shape=[4, 84, 84],
            action_shape=6,
            encoder_hidden_size_list=[128, 128, 256],
            actor_head_hidden_size=256,
            critic_head_hidden_size=256,
        ),
        learn=dict(
            batch_size=128,
            learning_rate=6e-4,
            grad_clip_type='clip_norm',
            clip_value=5,
            value_weight=0.5,
            entropy_weight=0.01,
            discount_factor=0.99,
            lambda_=0.95,
            bf16=bf16,
            # hipGraph-capture the learn step (single-process only)
            cuda_graph=os.environ.get('DING_IMPALA_GRAPH', '1') not in ('0', 'false'),
        ),
    )
    cfg = EasyDict(deep_merge_dicts(cfg, user))
    return _maybe_channels_last(IMPALAPolicy(cfg, enable_field=['learn']))


def impala_step(policy, device: str, batch_size: int = 128, unroll_len: int = 32):
    """One IMPALA learn iteration through IMPALAPolicy._forward_learn on a
    synthetic device-resident time-major batch (the same-node trajectory
    fast-path input format)."""
    T, B = unroll_len, batch_size
    obs_plus_1 = torch.rand(T + 1, B, 4, 84, 84, device=device)
    if _channels_last_on():
        obs_plus_1 = obs_plus_1.reshape((T + 1) * B, 4, 84, 84) \
            .to(memory_format=torch.channels_last).view(T + 1, B, 4, 84, 84)
    with torch.no_grad():
        sample = policy._model(obs_plus_1[0], mode='compute_actor')['logit']
    behaviour = sample.detach().unsqueeze(0).expand(T, B, -1).contiguous()
    behaviour = behaviour + 0.1 * torch.randn_like(behaviour)
    gumbel = -torch.log(-torch.log(torch.rand_like(behaviour) + 1e-10) + 1e-10)
    action = (behaviour + gumbel).argmax(dim=-1)
    batch = {
        'obs_plus_1': obs_plus_1,
        'logit': behaviour,
        'action': action,
        'reward': torch.randn(T, B, device=device),
        'done': (torch.rand(T, B, device=device) < 0.002).float(),
    }
    policy._forward_learn(batch)
    return T * B


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument('--gpus', type=int, default=1)
    parser.add_argument('--steps', type=int, default=5)
    parser.add_argument('--warmup', type=int, default=2)
    parser.add_argument('--workload', type=str, default='ppo', choices=['ppo', 'impala'])
    parser.add_argument('--n-sample', type=int, default=3200)
    parser.add_argument('--dtype', type=str, default='fp32', choices=['fp32', 'bf16'])
    args = parser.parse_args()

    world_size = int(os.environ.get('WORLD_SIZE', '1'))
    rank = int(os.environ.get('RANK', '0'))
    local_rank = int(os.environ.get('LOCAL_RANK', rank))
    distributed = world_size > 1

    use_gpu = torch.cuda.is_available()
    if use_gpu:
        torch.cuda.set_device(local_rank % max(1, torch.cuda.device_count()))
        device = f'cuda:{torch.cuda.current_device()}'
        if world_size > 1 and 'MIOPEN_USER_DB_PATH' not in os.environ:
            # per-rank find-db: 8 ranks sharing one sqlite user-db serialize
            # (or corrupt) the find-mode writes during warmup
            db = f'/tmp/miopen_udb_rank{local_rank}'
            os.makedirs(db, exist_ok=True)
            os.environ['MIOPEN_USER_DB_PATH'] = db
    else:
        device = 'cpu'

    if distributed:
        os.environ.setdefault('MASTER_ADDR', '127.0.0.1')
        os.environ.setdefault('MASTER_PORT', '29500')
        torch.distributed.init_process_group(backend='nccl' if use_gpu else 'gloo')

    torch.manual_seed(1234 + rank)

    if args.workload == 'ppo':
        policy = build_ppo_policy(device, multi_gpu=distributed, bf16=args.dtype == 'bf16')
        step_fn = lambda: ppo_step(policy, device, args.n_sample)
        model_name = 'pong_ppo(conv[64,64,128] 4x84x84)'
        config = {
            'model': model_name, 'global_batch': args.n_sample * world_size, 'seq_len': 1,
            'parallelism': f'dp{world_size}', 'minibatch': 320, 'epoch_per_collect': 10,
        }
    else:
        policy = build_impala(device, multi_gpu=distributed, bf16=args.dtype == 'bf16')
        step_fn = lambda: impala_step(policy, device)
        config = {
            'model': 'spaceinvaders_impala(conv[128,128,256] 4x84x84)', 'global_batch': 128 * 32 * world_size,
            'seq_len': 32, 'parallelism': f'dp{world_size}',
        }

    def barrier_sync():
        if distributed:
            torch.distributed.barrier()
        if use_gpu:
            torch.cuda.synchronize()

    # warmup
    for _ in range(args.warmup):
        step_fn()
    barrier_sync()

    t0 = time.perf_counter()
    samples = 0
    for _ in range(args.steps):
        samples += step_fn()
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if distributed:
        t = torch.tensor([elapsed], device=device if use_gpu else 'cpu')
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = t.item()

    total_samples = samples * world_size
    value = total_samples / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        print(json.dumps({
            'metric': 'learner samples/sec (env-steps/sec), Atari ' + args.workload.upper(),
            'value': value,
            'unit': 'samples/s',
            'n_gpus': world_size,
            'steps': args.steps,
            'warmup': args.warmup,
            'ms_per_step': ms_per_step,
            'higher_is_better': True,
            'scaling': 'weak',
            'vs_baseline': None,
            'dtype': args.dtype,
            'data': 'synthetic',
            'config': config,
        }))

    if distributed:
        torch.distributed.destroy_process_group()


if __name__ == '__main__':
    main()
