"""Bucketed gradient all-reduce for data-parallel training over RCCL/xGMI.

Replaces the reference's per-parameter async hooks
(opendilab/DI-engine ding/policy/base_policy.py:185-199, :415-460
sync_gradients) with an MI355X-native design:

* Parameters are packed into ~25 MB flat bf16/fp32 buckets in reverse
  parameter order (the order backward produces grads), so each RCCL
  all-reduce message is large enough to stripe across the 7 xGMI
  point-to-point links instead of hundreds of latency-bound tiny calls.
* In async mode the bucket all-reduce launches on a dedicated HIP stream as
  soon as the bucket's grads are ready, overlapping with the rest of
  backward; ``sync()`` waits and scatters the reduced flat buffer back.
* With ``with_indicator=True`` the reference's partially-used-network
  averaging (opendilab/DI-engine ding/policy/base_policy.py:415-460) is
  reproduced: each bucket carries one extra indicator float per parameter
  (1.0 if this rank produced a grad, else 0.0) appended to the flat buffer,
  so the participation counts ride in the SAME all-reduce message as the
  grads (no extra collective), and each parameter's reduced grad is divided
  by the number of participating ranks instead of world_size.
"""
from typing import Dict, List, Optional

import torch
import torch.distributed as dist


class GradBucket:

    def __init__(self, params: List[torch.nn.Parameter], flat: torch.Tensor, offsets: List[int]):
        self.params = params
        self.flat = flat
        self.offsets = offsets
        self.ready_count = 0
        self.ready = False
        self.work = None

    def reset(self):
        self.ready_count = 0
        self.ready = False
        self.work = None


class GradBucketAllReducer:

    def __init__(
        self,
        model: torch.nn.Module,
        bucket_bytes: int = 25 * 1024 * 1024,
        async_overlap: bool = True,
        with_indicator: bool = False,
    ):
        self.model = model
        self.bucket_bytes = bucket_bytes
        self.async_overlap = async_overlap and torch.cuda.is_available()
        self.with_indicator = with_indicator
        self._params = [p for p in model.parameters() if p.requires_grad]
        self._buckets: List[GradBucket] = []
        self._param_to_bucket: Dict[int, GradBucket] = {}
        self._comm_stream = torch.cuda.Stream() if torch.cuda.is_available() else None
        self._next_launch = 0
        self._build_buckets()
        if self.async_overlap:
            self._register_hooks()

    # ------------------------------------------------------------- building
    def _build_buckets(self):
        current: List[torch.nn.Parameter] = []
        current_numel = 0
        max_numel = self.bucket_bytes // 4

        def close():
            nonlocal current, current_numel
            if not current:
                return
            device = current[0].device
            # indicator mode: one trailing float per param carries this rank's
            # participation bit inside the same all-reduce message
            extra = len(current) if self.with_indicator else 0
            flat = torch.zeros(current_numel + extra, dtype=torch.float32, device=device)
            offsets, off = [], 0
            for p in current:
                offsets.append(off)
                off += p.numel()
            b = GradBucket(list(current), flat, offsets)
            self._buckets.append(b)
            for p in current:
                self._param_to_bucket[id(p)] = b
            current, current_numel = [], 0

        # reverse order approximates autograd completion order
        for p in reversed(self._params):
            current.append(p)
            current_numel += p.numel()
            if current_numel >= max_numel:
                close()
        close()

    def _register_hooks(self):
        for p in self._params:
            p.register_post_accumulate_grad_hook(self._on_grad_ready)

    # ------------------------------------------------------------- runtime
    def broadcast_params(self, src: int = 0):
        for p in self._params:
            dist.broadcast(p.data, src)

    def _launch(self, bucket: GradBucket):
        world = dist.get_world_size()
        # pack grads into the flat buffer
        ind_off = bucket.offsets[-1] + bucket.params[-1].numel() if self.with_indicator else None
        for j, (p, off) in enumerate(zip(bucket.params, bucket.offsets)):
            n = p.numel()
            if p.grad is not None:
                bucket.flat[off:off + n].copy_(p.grad.detach().reshape(-1))
            else:
                bucket.flat[off:off + n].zero_()
            if ind_off is not None:
                bucket.flat[ind_off + j] = 1.0 if p.grad is not None else 0.0
        if not self.with_indicator:
            bucket.flat.div_(world)
        if self._comm_stream is not None:
            self._comm_stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(self._comm_stream):
                bucket.work = dist.all_reduce(bucket.flat, async_op=True)
        else:
            bucket.work = dist.all_reduce(bucket.flat, async_op=True)

    def _on_grad_ready(self, param: torch.nn.Parameter):
        if not dist.is_available() or not dist.is_initialized():
            return
        bucket = self._param_to_bucket[id(param)]
        bucket.ready_count += 1
        if bucket.ready_count == len(bucket.params):
            bucket.ready = True
            # launch strictly in bucket order so every rank issues the same
            # all-reduce sequence even when parameter usage differs per rank
            self._drain_ready()

    def _drain_ready(self):
        while self._next_launch < len(self._buckets):
            b = self._buckets[self._next_launch]
            if not b.ready:
                break
            self._launch(b)
            self._next_launch += 1

    def sync(self):
        """Finish outstanding bucket reductions and write back grads. In
        synchronous mode this launches all buckets now."""
        if not dist.is_available() or not dist.is_initialized():
            return
        for bucket in self._buckets[self._next_launch:]:
            self._launch(bucket)
        self._next_launch = len(self._buckets)
        for bucket in self._buckets:
            bucket.work.wait()
        if self._comm_stream is not None:
            torch.cuda.current_stream().wait_stream(self._comm_stream)
        for bucket in self._buckets:
            counts = None
            if self.with_indicator:
                # divide each param's summed grad by its participation count
                ind_off = bucket.offsets[-1] + bucket.params[-1].numel()
                counts = bucket.flat[ind_off:ind_off + len(bucket.params)].clamp_min(1.0)
                for j, (p, off) in enumerate(zip(bucket.params, bucket.offsets)):
                    bucket.flat[off:off + p.numel()].div_(counts[j])
                raw_counts = bucket.flat[ind_off:ind_off + len(bucket.params)].cpu()
            for j, (p, off) in enumerate(zip(bucket.params, bucket.offsets)):
                n = p.numel()
                if p.grad is None:
                    if counts is not None and raw_counts[j] == 0.0:
                        continue  # no rank used this param: leave grad None
                    p.grad = bucket.flat[off:off + n].reshape(p.shape).clone()
                else:
                    p.grad.detach().reshape(-1).copy_(bucket.flat[off:off + n])
            bucket.reset()
        self._next_launch = 0


def sync_gradients_flat(model: torch.nn.Module, bucket_bytes: int = 25 * 1024 * 1024):
    """One-shot synchronous bucketed grad all-reduce (no persistent state)."""
    if not dist.is_available() or not dist.is_initialized():
        return
    world = dist.get_world_size()
    params = [p for p in model.parameters() if p.requires_grad and p.grad is not None]
    if not params:
        return
    max_numel = bucket_bytes // 4
    i = 0
    while i < len(params):
        chunk = []
        numel = 0
        while i < len(params) and numel < max_numel:
            chunk.append(params[i])
            numel += params[i].numel()
            i += 1
        flat = torch.cat([p.grad.detach().reshape(-1) for p in chunk])
        flat.div_(world)
        dist.all_reduce(flat)
        off = 0
        for p in chunk:
            n = p.numel()
            p.grad.detach().reshape(-1).copy_(flat[off:off + n])
            off += n
