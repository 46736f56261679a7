"""Adapter/Coordinator transport + serialization tests (reference:
ctools/worker/coordinator/{adapter,coordinator}.py)."""
import threading

import pytest
import torch

from distar_amd.data.adapter import Adapter
from distar_amd.data.coordinator import Coordinator
from distar_amd.utils.serialize import dumps, loads


def test_serialize_roundtrip_tensor_tree():
    data = {'a': torch.randn(4, 5), 'b': [torch.arange(3), {'c': torch.randn(2).bfloat16()}],
            'd': 'text', 'e': 7}
    for fs_type in ('pickle', 'nppickle', 'torch'):
        blob = dumps(data, fs_type=fs_type)
        out = loads(blob, fs_type=fs_type)
        torch.testing.assert_close(out['a'], data['a'])
        torch.testing.assert_close(out['b'][0], data['b'][0])
        torch.testing.assert_close(out['b'][1]['c'].float(), data['b'][1]['c'].float())
        assert out['d'] == 'text' and out['e'] == 7


@pytest.mark.timeout(60)
def test_adapter_push_pull_roundtrip():
    coord = Coordinator().run()
    try:
        producer = Adapter(coordinator_port=coord.port)
        consumer = Adapter(coordinator_port=coord.port)
        payloads = [{'step': torch.full((3,), float(i))} for i in range(4)]
        for p in payloads:
            producer.push(p, token='MP0traj')
        assert producer.length('MP0traj') == 4
        out = consumer.pull('MP0traj', size=4, worker_num=2, timeout=30)
        assert len(out) == 4
        got = sorted(float(d['step'][0]) for d in out)
        assert got == [0.0, 1.0, 2.0, 3.0]
        assert consumer.length('MP0traj') == 0
    finally:
        coord.close()


@pytest.mark.timeout(60)
def test_adapter_pull_blocks_until_data():
    coord = Coordinator().run()
    try:
        producer = Adapter(coordinator_port=coord.port)
        consumer = Adapter(coordinator_port=coord.port)
        result = []

        def late_push():
            import time
            time.sleep(0.5)
            producer.push({'x': torch.ones(2)}, token='t2')

        t = threading.Thread(target=late_push)
        t.start()
        out = consumer.pull('t2', size=1, sleep_time=0.1, timeout=30)
        t.join()
        assert len(out) == 1
        torch.testing.assert_close(out[0]['x'], torch.ones(2))
    finally:
        coord.close()


@pytest.mark.timeout(60)
def test_adapter_bounded_queue_drops_oldest():
    coord = Coordinator().run()
    try:
        producer = Adapter(coordinator_port=coord.port, maxlen=2)
        consumer = Adapter(coordinator_port=coord.port)
        for i in range(4):
            producer.push({'i': torch.tensor([i])}, token='t3')
        # coordinator still lists all 4 metadata entries, but only the 2
        # newest producers hold live sockets; stale ones fail + are skipped
        out = consumer.pull('t3', size=2, sleep_time=0.05, timeout=30)
        vals = sorted(int(d['i'][0]) for d in out)
        assert vals == [2, 3]
    finally:
        coord.close()


@pytest.mark.timeout(120)
def test_rl_dataloader_pulls_and_collates():
    """RLDataLoader end-to-end on CPU: trajectories pushed through the
    Adapter come out as collated learner batches."""
    import random
    from distar_amd.data.rl_dataloader import RLDataLoader
    from distar_amd.lib.fake_data import fake_obs_step, fake_rl_step
    from distar_amd.utils.config import Config
    torch.manual_seed(0)
    coord = Coordinator().run()
    loader = None
    try:
        producer = Adapter(coordinator_port=coord.port)
        rng = random.Random(0)
        for _ in range(3):
            steps = [fake_rl_step(48, rng) for _ in range(2)]
            last = fake_obs_step(entity_num=48)
            last['hidden_state'] = [(torch.zeros(384), torch.zeros(384))
                                    for _ in range(3)]
            for k in last['entity_info']:
                last['entity_info'][k] = last['entity_info'][k][:48]
            producer.push(steps + [last], token='MP0traj', fs_type='nppickle')
        cfg = Config({'learner': {'player_id': 'MP0',
                                  'data': {'batch_size': 2, 'buffer_size': 2,
                                           'use_async_cuda': False}},
                      'communication': {'coordinator_ip': '127.0.0.1',
                                        'coordinator_port': coord.port,
                                        'adapter_traj_worker_num': 1}})
        loader = RLDataLoader(cfg)
        batch = next(loader)
        assert batch['batch_size'] == 2 and batch['unroll_len'] == 2
        assert batch['entity_num'].shape[0] == (2 + 1) * 2   # (T+1)*B rows
        assert batch['action_info']['action_type'].shape == (2, 2)
    finally:
        if loader is not None:
            loader.close()
        coord.close()


@pytest.mark.timeout(300)
def test_model_publication_refreshes_actor_weights(tmp_path, monkeypatch):
    """Learner publishes a policy state_dict; the actor's pull_model applies
    it to the agent (reference learner_comm.py:53-99 + actor_comm.py:172-196).
    Value-network keys are stripped from the wire payload."""
    monkeypatch.chdir(tmp_path)
    from types import SimpleNamespace
    from distar_amd.actor.comm import ActorComm, LearnerComm
    from distar_amd.models import Model
    from distar_amd.utils.config import Config
    from distar_amd.utils.checkpoint import CountVar
    torch.manual_seed(0)
    coord = Coordinator().run()
    try:
        cfg = Config({'learner': {'player_id': 'MP0'},
                      'communication': {'coordinator_ip': '127.0.0.1',
                                        'coordinator_port': coord.port}})
        lcomm = LearnerComm(cfg)
        learner_model = Model(Config({}), use_value_network=True)
        fake_learner = SimpleNamespace(model=learner_model,
                                       last_iter=CountVar(123))
        lcomm.send_model(fake_learner)
        acomm = ActorComm(cfg)
        payload = acomm.pull_model('MP0', timeout=30)
        assert payload is not None and payload['model_last_iter'] == 123
        assert not any(k.startswith('value_networks') for k in payload['model'])
        actor_model = Model(Config({}))
        with torch.no_grad():
            for p in actor_model.parameters():
                p.zero_()
        actor_model.load_state_dict(payload['model'], strict=False)
        k = next(k for k in payload['model'])
        torch.testing.assert_close(actor_model.state_dict()[k],
                                   learner_model.state_dict()[k])
    finally:
        coord.close()


@pytest.mark.timeout(120)
def test_slow_consumer_still_gets_fresh_data():
    """A consumer far behind a bounded producer must still make progress:
    newest-first metadata hand-out means the fetched payload is always
    live (regression test for the oldest-first eviction livelock)."""
    coord = Coordinator().run()
    try:
        producer = Adapter(coordinator_port=coord.port, maxlen=4)
        consumer = Adapter(coordinator_port=coord.port)
        # producer races 50 pushes ahead of a 4-deep payload buffer
        for i in range(50):
            producer.push({'i': torch.tensor([i])}, token='fresh')
        out = consumer.pull('fresh', size=2, sleep_time=0.05, timeout=60)
        vals = [int(d['i'][0]) for d in out]
        assert len(vals) == 2
        assert all(v >= 46 for v in vals), vals    # only the live tail
    finally:
        coord.close()


@pytest.mark.timeout(120)
def test_coordinator_worker_offload():
    """start_worker spawns a per-token broker with the identical API
    (reference coordinator.py:20-59,156-165); adapters pointed at the worker
    port roundtrip normally."""
    from distar_amd.utils.http import post_json
    coord = Coordinator().run()
    worker_srv = None
    try:
        resp = post_json(f'http://127.0.0.1:{coord.port}/start_worker',
                         {'token': 'hot'})
        assert resp['port'] and resp['port'] != coord.port
        # same worker is reused per token
        again = post_json(f'http://127.0.0.1:{coord.port}/start_worker',
                          {'token': 'hot'})
        assert again['port'] == resp['port']
        producer = Adapter(coordinator_port=resp['port'])
        consumer = Adapter(coordinator_port=resp['port'])
        producer.push({'v': torch.tensor([7.0])}, token='hot')
        out = consumer.pull('hot', size=1, timeout=30)
        assert float(out[0]['v'][0]) == 7.0
    finally:
        coord.close()
