"""Distributed backend over torch.distributed (RCCL on ROCm, gloo on CPU).

Functional parity with the reference's `ctools/utils/dist_helper.py:221-440`
(dist_init flavors, allreduce with mean, broadcast, group split, no-op
fallback when not initialized).  The gradient-sync strategy is NOT the
reference's per-parameter synchronous loop — see `distar_amd/parallel/ddp.py`
for the xGMI-sized bucketed, backward-overlapped replacement.
"""
import os
import socket

import torch
import torch.distributed as dist


def is_initialized():
    return dist.is_available() and dist.is_initialized()


def get_rank():
    return dist.get_rank() if is_initialized() else 0


def get_world_size():
    return dist.get_world_size() if is_initialized() else 1


def allreduce(tensor, average=True, group=None):
    """SUM allreduce, divided by world size when ``average`` (reference
    `dist_helper.allreduce`).  No-op when single-process."""
    if not is_initialized():
        return tensor
    dist.all_reduce(tensor, group=group)
    if average:
        tensor.div_(dist.get_world_size(group=group))
    return tensor


def allreduce_async(tensor, group=None):
    if not is_initialized():
        return None
    return dist.all_reduce(tensor, group=group, async_op=True)


def broadcast(tensor, src=0, group=None):
    if not is_initialized():
        return tensor
    dist.broadcast(tensor, src=src, group=group)
    return tensor


def barrier():
    if is_initialized():
        dist.barrier()


def _free_port():
    s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    s.bind(('127.0.0.1', 0))
    port = s.getsockname()[1]
    s.close()
    return port


def dist_init(method='torch', rank=None, world_size=None, init_method=None,
              backend=None, device_id=None):
    """Initialize the process group.

    method:
      - 'torch'      rank/world_size/init_method given explicitly or via env
                     (torchrun sets RANK/WORLD_SIZE/MASTER_ADDR/MASTER_PORT)
      - 'single_node' spawn-style with a tcp:// init method
      - 'slurm'      derive rank/world from SLURM_* env
    Backend defaults to nccl (=RCCL) when a GPU is visible, else gloo.
    """
    if backend is None:
        backend = 'nccl' if torch.cuda.is_available() else 'gloo'
    if method == 'slurm':
        proc_id = int(os.environ['SLURM_PROCID'])
        ntasks = int(os.environ['SLURM_NTASKS'])
        node_list = os.environ['SLURM_NODELIST']
        addr = node_list[8:].replace('-', '.') if node_list.startswith('SLURM') else node_list
        os.environ.setdefault('MASTER_ADDR', addr)
        os.environ.setdefault('MASTER_PORT', '29500')
        rank, world_size = proc_id, ntasks
        dist.init_process_group(backend=backend, rank=rank, world_size=world_size)
    elif method == 'single_node':
        assert rank is not None and world_size is not None
        init_method = init_method or f'tcp://127.0.0.1:{_free_port()}'
        dist.init_process_group(backend=backend, init_method=init_method,
                                rank=rank, world_size=world_size)
    else:  # 'torch'
        if rank is None and 'RANK' in os.environ:
            rank = int(os.environ['RANK'])
        if world_size is None and 'WORLD_SIZE' in os.environ:
            world_size = int(os.environ['WORLD_SIZE'])
        kwargs = {}
        if init_method is not None:
            kwargs['init_method'] = init_method
        dist.init_process_group(backend=backend, rank=rank or 0,
                                world_size=world_size or 1, **kwargs)
    if torch.cuda.is_available():
        local_rank = int(os.environ.get('LOCAL_RANK', get_rank() % max(torch.cuda.device_count(), 1)))
        torch.cuda.set_device(local_rank)
    return get_rank(), get_world_size()


def dist_finalize():
    if is_initialized():
        dist.destroy_process_group()


def simple_group_split(world_size, rank, num_groups):
    """Split the world into ``num_groups`` equal groups; return this rank's
    group handle (reference `dist_helper.simple_group_split`)."""
    groups = []
    rank_list = torch.arange(world_size).split(world_size // num_groups)
    for ranks in rank_list:
        groups.append(dist.new_group(ranks.tolist()))
    return groups[rank // (world_size // num_groups)]
