"""Same-node actor->learner IMPALA with GPU trajectory shipping.

Rank 0 (actor) builds device-resident unroll batches and ships them over
RCCL/xGMI via the gpu-exchanger middleware (header-only pickle, flat
per-dtype dist.send); rank 1 (learner) feeds them straight into
IMPALAPolicy's collated-batch fast path — the trajectory tensors never
visit host memory on a real node.

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \\
        --master-addr 127.0.0.1 dizoo/atari/example/atari_impala_gpu_shipper.py
"""
import os

import torch


def main(iters: int = 10, unroll_len: int = 32, batch_size: int = 16):
    import torch.distributed as dist
    from ding.framework.middleware import gpu_trajectory_sender, gpu_trajectory_receiver
    from ding.policy import IMPALAPolicy
    from ding.utils import EasyDict, deep_merge_dicts

    rank = int(os.environ.get('RANK', '0'))
    use_gpu = torch.cuda.is_available()
    if use_gpu:
        torch.cuda.set_device(int(os.environ.get('LOCAL_RANK', rank)))
    dist.init_process_group('nccl' if use_gpu else 'gloo')
    device = f"cuda:{torch.cuda.current_device()}" if use_gpu else 'cpu'
    T, B, N = unroll_len, batch_size, 6

    if rank == 0:  # actor: synthesize device-resident unrolls and ship them
        send = gpu_trajectory_sender(dst=1, collate=False)

        class Ctx:
            trajectories = None

        for it in range(iters):
            behaviour = torch.randn(T, B, N, device=device)
            ctx = Ctx()
            ctx.env_step = (it + 1) * T * B
            ctx.train_data = {
                'obs_plus_1': torch.rand(T + 1, B, 4, 84, 84, device=device),
                'logit': behaviour,
                'action': behaviour.argmax(-1),
                'reward': torch.randn(T, B, device=device),
                'done': torch.zeros(T, B, device=device),
            }
            send(ctx)
        print('[actor] shipped', iters, 'unroll batches')
    else:  # learner
        cfg = EasyDict(deep_merge_dicts(IMPALAPolicy.default_config(), EasyDict(dict(
            cuda=use_gpu,
            model=dict(obs_shape=[4, 84, 84], action_shape=N, encoder_hidden_size_list=[64, 64, 128]),
            learn=dict(batch_size=B),
        ))))
        policy = IMPALAPolicy(cfg, enable_field=['learn'])
        recv = gpu_trajectory_receiver(src=0, device=device)

        class Ctx:
            env_step = 0
            train_data = None

        ctx = Ctx()
        for it in range(iters):
            recv(ctx)
            out = policy._forward_learn(ctx.train_data)
            print(f"[learner] iter {it} env_step {ctx.env_step} loss {out['total_loss']:.4f}")
    dist.destroy_process_group()


if __name__ == '__main__':
    main()
