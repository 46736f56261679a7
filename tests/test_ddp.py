"""Bucketed overlapped allreduce (DistModule) correctness over gloo,
world_size=2, CPU (the GPU path is the same code over RCCL)."""
import os
import tempfile

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _worker(rank, world, init_file, q):
    dist.init_process_group('gloo', init_method=f'file://{init_file}',
                            rank=rank, world_size=world)
    from distar_amd.parallel.ddp import DistModule
    torch.manual_seed(7)          # same init on both ranks pre-broadcast
    model = torch.nn.Sequential(
        torch.nn.Linear(16, 32), torch.nn.ReLU(), torch.nn.Linear(32, 4))
    # a parameter that never gets a grad (frozen-branch case)
    model.extra = torch.nn.Linear(4, 4)
    dm = DistModule(model, bucket_cap_mb=1)
    torch.manual_seed(100 + rank)  # different data per rank
    x = torch.randn(8, 16)
    y = dm.module[2](torch.relu(dm.module[0](x))).sum()
    y.backward()
    dm.sync_gradients()
    # send plain numpy copies: mp.Queue ships torch tensors via fd-passed
    # shared memory, which breaks if the child exits before the parent
    # materializes them (ConnectionResetError in recvfds)
    grads = {n: p.grad.clone().numpy() for n, p in model.named_parameters()
             if p.grad is not None}
    q.put((rank, grads))
    dist.barrier()
    dist.destroy_process_group()


def _run_workers():
    with tempfile.TemporaryDirectory() as d:
        init_file = os.path.join(d, 'init')
        ctx = mp.get_context('spawn')
        q = ctx.Queue()
        procs = [ctx.Process(target=_worker, args=(r, 2, init_file, q))
                 for r in range(2)]
        for p in procs:
            p.start()
        results = {}
        for _ in range(2):
            rank, grads = q.get(timeout=90)
            results[rank] = grads
        for p in procs:
            p.join(timeout=60)
    return results


@pytest.mark.timeout(300)
def test_distmodule_grads_are_averaged():
    # spawn + file-store rendezvous can flake under a loaded test host;
    # bounded retries keep the signal without masking real failures
    import time as _time
    for attempt in range(3):
        try:
            results = _run_workers()
            break
        except Exception:
            if attempt == 2:
                raise
            _time.sleep(5)
    # both ranks end with identical (averaged) gradients
    results = {r: {n: torch.from_numpy(g) for n, g in d.items()}
               for r, d in results.items()}
    assert set(results[0].keys()) == set(results[1].keys())
    for n in results[0]:
        torch.testing.assert_close(results[0][n], results[1][n])

    # reproduce the expected average single-process
    torch.manual_seed(7)
    model = torch.nn.Sequential(
        torch.nn.Linear(16, 32), torch.nn.ReLU(), torch.nn.Linear(32, 4))
    expected = {}
    for rank in range(2):
        torch.manual_seed(100 + rank)
        x = torch.randn(8, 16)
        model.zero_grad()
        model(x).sum().backward()
        for n, p in model.named_parameters():
            expected[n] = expected.get(n, 0) + p.grad / 2
    for n, g in expected.items():
        torch.testing.assert_close(results[0][n], g)


def _accum_worker(rank, world, init_file, q):
    dist.init_process_group('gloo', init_method=f'file://{init_file}',
                            rank=rank, world_size=world)
    from distar_amd.parallel.ddp import DistModule
    torch.manual_seed(7)
    model = torch.nn.Sequential(torch.nn.Linear(16, 8), torch.nn.Linear(8, 4))
    dm = DistModule(model, bucket_cap_mb=1)
    torch.manual_seed(100 + rank)
    # two backwards (gradient accumulation) before one sync: the eager bucket
    # launch after backward #1 must be superseded by the accumulated grads
    for _ in range(2):
        x = torch.randn(8, 16)
        dm.module(x).sum().backward()
    dm.sync_gradients()
    grads = {n: p.grad.clone().numpy() for n, p in model.named_parameters()}
    q.put((rank, grads))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_distmodule_gradient_accumulation():
    import time as _time
    for attempt in range(3):
        try:
            with tempfile.TemporaryDirectory() as d:
                init_file = os.path.join(d, 'init')
                ctx = mp.get_context('spawn')
                q = ctx.Queue()
                procs = [ctx.Process(target=_accum_worker,
                                     args=(r, 2, init_file, q))
                         for r in range(2)]
                for p in procs:
                    p.start()
                results = {}
                for _ in range(2):
                    rank, grads = q.get(timeout=90)
                    results[rank] = grads
                for p in procs:
                    p.join(timeout=60)
            break
        except Exception:
            if attempt == 2:
                raise
            _time.sleep(5)
    results = {r: {n: torch.from_numpy(g) for n, g in d.items()}
               for r, d in results.items()}
    # expected: average over ranks of the two-backward accumulated grads
    torch.manual_seed(7)
    model = torch.nn.Sequential(torch.nn.Linear(16, 8), torch.nn.Linear(8, 4))
    expected = {}
    for rank in range(2):
        torch.manual_seed(100 + rank)
        model.zero_grad()
        for _ in range(2):
            x = torch.randn(8, 16)
            model(x).sum().backward()
        for n, p in model.named_parameters():
            expected[n] = expected.get(n, 0) + p.grad / 2
    for n, g in expected.items():
        torch.testing.assert_close(results[0][n], g)
        torch.testing.assert_close(results[1][n], g)


def _cross_rank_worker(rank, world, init_file, q):
    dist.init_process_group('gloo', init_method=f'file://{init_file}',
                            rank=rank, world_size=world)
    from distar_amd.lib.fake_data import fake_sl_batch_fast
    from distar_amd.losses.sl_loss import SupervisedLoss
    from distar_amd.models.alphastar.model import Model
    from distar_amd.utils.config import Config
    torch.manual_seed(5)                  # same weights on both ranks
    model = Model(Config({}), temperature=1.0)
    torch.manual_seed(50 + rank)          # different data per rank
    data = fake_sl_batch_fast(batch_size=2, traj_len=2, seed=50 + rank)
    hidden = [(torch.zeros(2, 384), torch.zeros(2, 384)) for _ in range(3)]
    logits, infer, _ = model.sl_train(**data, hidden_state=hidden)
    loss = SupervisedLoss(Config({'learner': {'cross_rank_loss': True}}))
    ld = loss.compute_loss(logits, data['action_info'], data['action_mask'],
                           data['selected_units_num'], data['entity_num'],
                           infer)
    q.put((rank, float(loss.total_batch_size), float(ld['total_loss'])))
    dist.barrier()
    dist.destroy_process_group()


def _spawn_two(worker):
    """Run a 2-rank gloo worker with bounded retries (spawn + file-store
    rendezvous flakes under load)."""
    import time as _time
    last = None
    for attempt in range(3):
        try:
            with tempfile.TemporaryDirectory() as d:
                init_file = os.path.join(d, 'init')
                ctx = mp.get_context('spawn')
                q = ctx.Queue()
                procs = [ctx.Process(target=worker, args=(r, 2, init_file, q))
                         for r in range(2)]
                for p in procs:
                    p.start()
                out = {}
                for _ in range(2):
                    rank, *vals = q.get(timeout=240)
                    out[rank] = tuple(vals)
                for p in procs:
                    p.join(timeout=60)
                return out
        except Exception as e:  # noqa: BLE001
            last = e
            _time.sleep(5)
    raise last


@pytest.mark.timeout(900)
def test_sl_cross_rank_loss_renormalizes():
    """cross_rank_loss allreduces the global batch size and renormalizes
    each rank's loss by its share (reference sl_loss.py:100-104,127-135)."""
    out = {r: v for r, v in _spawn_two(_cross_rank_worker).items()}
    # (B*T)=4 rows per rank -> global 8 on both ranks
    assert out[0][0] == 8.0 and out[1][0] == 8.0
    assert out[0][1] > 0 and out[1][1] > 0


def _log_reduce_worker(rank, world, init_file, q):
    dist.init_process_group('gloo', init_method=f'file://{init_file}',
                            rank=rank, world_size=world)
    from distar_amd.learner.hooks import LogReduceHook
    from types import SimpleNamespace
    engine = SimpleNamespace(log_buffer={'loss': float(rank), 'acc': 1.0 + rank,
                                         'name': 'text-untouched'})
    LogReduceHook(position='after_iter')(engine)
    q.put((rank, dict(engine.log_buffer)))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(900)
def test_log_reduce_hook_averages_buffer():
    """LogReduceHook allreduce-averages every scalar in the log buffer as one
    flat tensor (reference learner_hook.py:271-325)."""
    out = {r: v[0] for r, v in _spawn_two(_log_reduce_worker).items()}
    for rank in (0, 1):
        assert out[rank]['loss'] == 0.5          # mean(0, 1)
        assert out[rank]['acc'] == 1.5           # mean(1, 2)
        assert out[rank]['name'] == 'text-untouched'
