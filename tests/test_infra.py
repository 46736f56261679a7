"""Learner-hook registry, timers, scalar logging (reference
`ctools/worker/learner/learner_hook.py`, `ctools/utils/time_helper.py`,
`ctools/utils/log_helper.py`)."""
import json
import os
import time
from types import SimpleNamespace

import torch

from distar_amd.learner.hooks import (Hook, add_learner_hook,
                                      build_learner_hook_by_cfg)
from distar_amd.utils.log import ScalarLogger
from distar_amd.utils.timing import EasyTimer


class _Probe(Hook):
    calls = []

    def __init__(self, tag, **kwargs):
        super().__init__(tag, **kwargs)
        self.tag = tag

    def __call__(self, engine):
        _Probe.calls.append(self.tag)


def test_hooks_run_in_priority_order():
    """Lower priority number runs first within a position (reference
    learner_hook.py registry semantics)."""
    _Probe.calls = []
    hooks = {p: [] for p in ('before_run', 'before_iter', 'after_iter',
                             'after_run')}
    add_learner_hook(hooks, _Probe('late', priority=90, position='after_iter'))
    add_learner_hook(hooks, _Probe('early', priority=10, position='after_iter'))
    add_learner_hook(hooks, _Probe('mid', priority=50, position='after_iter'))
    engine = SimpleNamespace()
    for h in hooks['after_iter']:
        h(engine)
    assert _Probe.calls == ['early', 'mid', 'late']


def test_build_hooks_from_cfg():
    from distar_amd.utils.config import Config
    cfg = Config({'after_iter': {'log_show': {'ext_args': {'freq': 5}}},
                  'before_run': {'load_ckpt': {'ext_args': {}}}})
    hooks = build_learner_hook_by_cfg(cfg)
    assert any(h.name == 'log_show' for h in hooks['after_iter'])
    assert any(h.name == 'load_ckpt' for h in hooks['before_run'])


def test_easy_timer_cpu():
    t = EasyTimer(cuda=False)
    with t:
        time.sleep(0.05)
    assert 0.04 < t.value < 1.0


def test_scalar_logger_jsonl(tmp_path):
    logger = ScalarLogger(str(tmp_path), name='test')
    logger.register_var('loss')
    logger.add_scalar('loss', 1.5, global_step=3)
    logger.add_scalar('loss', torch.tensor(2.5), global_step=4)
    logger.flush()
    lines = [json.loads(l) for l in
             open(os.path.join(tmp_path, 'test.jsonl'))]
    assert lines[0] == {**lines[0], 'step': 3, 'key': 'loss', 'value': 1.5}
    assert lines[1]['value'] == 2.5
    logger.close()


def test_stopwatch_hierarchy_and_report():
    from distar_amd.utils.timing import Stopwatch
    s = Stopwatch(enabled=True)
    for _ in range(3):
        with s('outer'):
            time.sleep(0.01)
            with s('inner'):
                time.sleep(0.01)
    report = str(s)
    assert 'outer' in report and 'outer.inner' in report
    assert s._times['outer'][0] == 3 and s._times['outer.inner'][0] == 3
    assert s._times['outer'][1] >= s._times['outer.inner'][1]
    # disabled stopwatch records nothing
    s2 = Stopwatch(enabled=False)
    with s2('x'):
        pass
    assert not s2._times

    @s.decorate('deco')
    def f():
        return 41 + 1
    assert f() == 42 and 'deco' in s._times


def test_seed_everything_reproducible_training():
    """Two seeded SL steps produce bit-identical losses (deterministic mode,
    SURVEY §5.2 rebuild requirement)."""
    from distar_amd.lib.fake_data import fake_sl_batch_fast
    from distar_amd.losses import SupervisedLoss
    from distar_amd.models import Model
    from distar_amd.utils.config import Config
    from distar_amd.utils.misc import seed_everything

    def one_loss():
        seed_everything(1234)
        m = Model(Config({}))
        data = fake_sl_batch_fast(batch_size=2, traj_len=2)
        hidden = [(torch.zeros(2, 384), torch.zeros(2, 384)) for _ in range(3)]
        logits, infer, _ = m.sl_train(**data, hidden_state=hidden)
        ld = SupervisedLoss(Config({'learner': {}})).compute_loss(
            logits, data['action_info'], data['action_mask'],
            data['selected_units_num'], data['entity_num'], infer)
        return float(ld['total_loss'])

    assert one_loss() == one_loss()


def test_alphastar_var_record_grid():
    """The RL metric grid renders registered {field}x{column} cells
    (reference log_helper.py:689-751)."""
    from distar_amd.utils.log import AlphaStarVarRecord
    rec = AlphaStarVarRecord(length=4)
    for key in ('winloss/reward', 'winloss/td', 'battle/value', 'kl/action_type'):
        rec.register_var(key)
        rec.update_var({key: 1.25})
    text = rec.get_star_text()
    assert 'winloss' in text and 'battle' in text and 'kl' in text
    assert '1.25000' in text
    # pretty_print survives nested dicts
    from distar_amd.utils.log import pretty_print
    out = pretty_print({'a': {'b': 1, 'c': 'x'}, 'd': 2.5}, direct_print=False)
    assert 'b' in out and 'd' in out
