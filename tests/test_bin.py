"""Entry-point smoke tests: play on the mock env, gen_z over pre-decoded
data, sl_train argument plumbing."""
import json
import os
import sys

import pytest
import torch


@pytest.mark.timeout(900)
def test_play_mock(tmp_path, monkeypatch, capsys):
    monkeypatch.chdir(tmp_path)
    from distar_amd.bin.play import main
    main(['--env', 'mock', '--episodes', '1', '--config', 'nonexistent.yaml'])
    out = capsys.readouterr().out
    assert 'winloss' in out or '0' in out


def test_gen_z_offline(tmp_path):
    from distar_amd.bin.gen_z import main
    from distar_amd.lib.actions import BEGINNING_ORDER_ACTIONS
    data_dir = tmp_path / 'decoded'
    data_dir.mkdir()
    traj = [{'action_info': {
        'action_type': torch.tensor(BEGINNING_ORDER_ACTIONS[5]),
        'target_location': torch.tensor(1234)}}]
    torch.save({'traj_data': traj, 'map_name': 'KingsCove', 'race': 'zerg',
                'opponent_race': 'zerg', 'born_location': 777, 'end_loop': 5000},
               data_dir / 'r0.pt')
    out = tmp_path / 'z.json'
    main(['--data', str(data_dir), '--output', str(out)])
    z = json.load(open(out))
    entry = z['KingsCove']['zerg']['777'][0]
    assert entry[0][0] == 5 and entry[2][0] == 1234 and entry[3] == 5000


def test_sl_train_args():
    from distar_amd.bin.sl_train import get_args
    args = get_args(['--type', 'learner', '--world-size', '4', '--rank', '1'])
    assert args.world_size == 4 and args.rank == 1


@pytest.mark.timeout(900)
def test_bench_distributed_cpu_gloo(tmp_path):
    """The driver runs bench.py under torch.distributed.run at N=1,2,4,8 on
    the GPU node; validate the whole multi-rank path (dist_init, DistModule
    buckets, MAX-over-ranks reduce, single JSON line from rank 0) on CPU with
    gloo, world_size=2, tiny shapes."""
    import json as _json
    import subprocess
    import sys
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    port = 29000 + os.getpid() % 1000
    cmd = [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
           '--nproc-per-node', '2', '--master-addr', '127.0.0.1',
           '--master-port', str(port),
           os.path.join(repo, 'bench.py'), '--gpus', '2', '--steps', '1',
           '--warmup', '0', '--batch', '2', '--traj', '2', '--pool', '1',
           '--mode', 'sl']
    env = dict(os.environ)
    env['OMP_NUM_THREADS'] = '2'
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=800,
                         cwd=str(tmp_path), env=env)
    assert out.returncode == 0, out.stderr[-2000:]
    json_lines = [l for l in out.stdout.splitlines() if l.startswith('{')]
    assert len(json_lines) == 1, out.stdout
    res = _json.loads(json_lines[0])
    assert res['n_gpus'] == 2 and res['steps'] == 1
    assert res['config']['parallelism'] == 'dp2'
    assert res['config']['global_batch'] == 4
    assert res['value'] > 0 and res['ms_per_step'] > 0


@pytest.mark.timeout(900)
def test_bench_distributed_cpu_gloo_rl(tmp_path):
    """Same as above for the RL bench path (value-feature critic model)."""
    import json as _json
    import subprocess
    import sys
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    port = 29000 + (os.getpid() + 7) % 1000
    cmd = [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
           '--nproc-per-node', '2', '--master-addr', '127.0.0.1',
           '--master-port', str(port),
           os.path.join(repo, 'bench.py'), '--gpus', '2', '--steps', '1',
           '--warmup', '0', '--batch', '2', '--traj', '2', '--pool', '1',
           '--entities', '64', '--mode', 'rl']
    env = dict(os.environ)
    env['OMP_NUM_THREADS'] = '2'
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=800,
                         cwd=str(tmp_path), env=env)
    assert out.returncode == 0, out.stderr[-2000:]
    json_lines = [l for l in out.stdout.splitlines() if l.startswith('{')]
    assert len(json_lines) == 1, out.stdout
    res = _json.loads(json_lines[0])
    assert res['n_gpus'] == 2 and res['config']['mode'] == 'rl'
    assert res['config']['value_feature'] is True


@pytest.mark.timeout(600)
def test_play_loads_model_checkpoints(tmp_path, monkeypatch):
    """play/eval without a league: actor loads actor.model{0,1}_path into the
    agents (reference play.py model_paths contract)."""
    monkeypatch.chdir(tmp_path)
    from distar_amd.actor.actor import Actor
    from distar_amd.models import Model
    from distar_amd.utils.checkpoint import CheckpointHelper
    from distar_amd.utils.config import Config
    torch.manual_seed(0)
    m = Model(Config({}))
    ckpt = str(tmp_path / 'custom.pth')
    CheckpointHelper().save(ckpt, m)
    cfg = Config({'actor': {'episode_num': 1, 'env_type': 'mock',
                            'traj_len': 4, 'job_type': 'eval_test',
                            'model0_path': ckpt},
                  'env': {'player_num': 2, 'max_episode_steps': 3},
                  'common': {'experiment_name': 'test_play_load',
                             'type': 'play'}})
    actor = Actor(cfg)
    actor._setup_job()
    a0 = actor._agents[0]
    sd = m.state_dict()
    got = a0.model.state_dict()
    k = next(iter(sd))
    torch.testing.assert_close(got[k], sd[k])


@pytest.mark.timeout(600)
def test_play_checkpoint_strategy_context(tmp_path, monkeypatch):
    """Play checkpoints carrying map_name/z_path/fake_reward_prob/z_idx apply
    to the loaded agent (reference actor.py:65-73, agent.py:202-204)."""
    monkeypatch.chdir(tmp_path)
    from distar_amd.actor.actor import Actor
    from distar_amd.models import Model
    from distar_amd.utils.config import Config
    torch.manual_seed(0)
    m = Model(Config({}))
    ckpt = str(tmp_path / 'play.pth')
    torch.save({'model': m.state_dict(), 'map_name': 'NewRepugnancy',
                'z_path': '7map_filter_spine.json', 'fake_reward_prob': 0.25,
                'z_idx': None}, ckpt)
    cfg = Config({'actor': {'episode_num': 1, 'env_type': 'mock',
                            'traj_len': 4, 'job_type': 'eval_test',
                            'model0_path': ckpt},
                  'env': {'player_num': 2, 'max_episode_steps': 3},
                  'common': {'experiment_name': 'test_play_ctx',
                             'type': 'play'}})
    actor = Actor(cfg)
    job = actor._setup_job()
    a0 = actor._agents[0]
    assert job['env_info']['map_name'] == 'NewRepugnancy'
    assert a0._cfg['z_path'] == '7map_filter_spine.json'
    assert a0._cfg['fake_reward_prob'] == 0.25
