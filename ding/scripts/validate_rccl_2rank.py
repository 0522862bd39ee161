"""Validate the RCCL codepath with 2 ranks (CI helper): bucketed indicator
all-reduce stays in sync and the trajectory shipper moves GPU tensors
rank-to-rank. Run via torchrun on a box with >= 2 GPUs — RCCL (like NCCL)
refuses two ranks on the same device ("Duplicate GPU detected", verified on
a 1-GPU MI355X box 2026-09-12), so each rank binds cuda:LOCAL_RANK. The
same reducer/shipper logic is covered on CPU by the gloo world_size=2
tests in tests/test_distributed.py. See bench_scale.sh for the full
per-GPU scaling sweep."""
import os
import sys

# torchrun children get the script's dir, not the repo root, on sys.path
sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), '..', '..')))

import torch
import torch.distributed as dist


def main():
    rank = int(os.environ['RANK'])
    local = int(os.environ.get('LOCAL_RANK', rank))
    if torch.cuda.device_count() < int(os.environ.get('WORLD_SIZE', 2)):
        raise SystemExit('needs one GPU per rank: RCCL rejects two ranks on one device')
    torch.cuda.set_device(local)
    dist.init_process_group('nccl')
    torch.manual_seed(100 + rank)
    # 1) bucketed reducer with indicator
    from ding.parallel import GradBucketAllReducer
    m = torch.nn.Sequential(torch.nn.Linear(64, 64), torch.nn.ReLU(), torch.nn.Linear(64, 8)).cuda()
    red = GradBucketAllReducer(m, bucket_bytes=4096, async_overlap=True, with_indicator=True)
    red.broadcast_params(src=0)
    for _ in range(3):
        for p in m.parameters():
            p.grad = None
        x = torch.randn(16, 64, device='cuda') * (rank + 1)
        m(x).pow(2).mean().backward()
        red.sync()
    g = next(m.parameters()).grad.sum()
    gs = [torch.zeros_like(g) for _ in range(2)]
    dist.all_gather(gs, g)
    assert torch.allclose(gs[0], gs[1], atol=1e-5), "grads out of sync over RCCL"
    # 2) trajectory shipper GPU->GPU
    from ding.data import TrajectoryShipper
    ship = TrajectoryShipper()
    if rank == 0:
        batch = {'obs': torch.arange(3 * 4 * 6, dtype=torch.float32, device='cuda').reshape(3, 4, 6),
                 'act': torch.randint(0, 4, (3, 4), device='cuda')}
        ship.send(batch, dst=1)
    else:
        got = ship.recv(src=0)
        assert got['obs'].is_cuda and got['obs'].shape == (3, 4, 6)
        assert float(got['obs'].sum()) == float(torch.arange(72).sum())
    dist.barrier()
    if rank == 0:
        print('RCCL_2RANK_OK')
    dist.destroy_process_group()


if __name__ == '__main__':
    main()
