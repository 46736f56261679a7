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
