"""Core-utility parity tests: metrics (reference `ctools/torch_utils/metric.py`),
grad-clip family (`ctools/torch_utils/grad_clip.py`), checkpoint helper
(`ctools/utils/checkpoint_helper.py`), log records (`ctools/utils/log_helper.py`)."""
import torch

from distar_amd.utils.checkpoint import CheckpointHelper, CountVar
from distar_amd.utils.grad_clip import GradClip
from distar_amd.utils.metric import (hamming_distance, l2_distance,
                                     levenshtein_distance)


# ------------------------------------------------------------------- metrics

def test_levenshtein_basic():
    a = torch.tensor([1, 2, 3, 4])
    b = torch.tensor([1, 3, 4, 5])
    # delete 2, insert 5 -> distance 2
    assert float(levenshtein_distance(a, b)) == 2.0
    assert float(levenshtein_distance(a, a)) == 0.0
    assert float(levenshtein_distance(a[:0], b)) == 4.0


def test_levenshtein_location_cost():
    """Equal build-order actions still pay a distance-scaled location cost
    (reference metric.py:14-60 extra_fn; used by the build-order pseudo-reward
    agent.py update_fake_reward)."""
    a = torch.tensor([7, 8])
    b = torch.tensor([7, 8])
    loc_a = torch.tensor([0, 0])                      # top-left corner
    loc_b = torch.tensor([0, 159 + 160 * 151])        # opposite corner
    d = levenshtein_distance(a, b, loc_a, loc_b,
                             extra_fn=lambda x, y: l2_distance(x, y))
    assert float(levenshtein_distance(a, b)) == 0.0
    assert 0.0 < float(d) <= 2 * 0.8                  # capped at max_val each


def test_hamming_and_l2():
    p = torch.tensor([[1, 0, 1, 1], [0, 0, 0, 0]])
    t = torch.tensor([[1, 1, 1, 0], [0, 0, 0, 0]])
    assert hamming_distance(p, t).tolist() == [2.0, 0.0]
    same = l2_distance(torch.tensor(5), torch.tensor(5))
    far = l2_distance(torch.tensor(0), torch.tensor(159 + 160 * 151))
    assert float(same) == 0.0 and abs(float(far) - 0.8) < 1e-6   # clamped


# ----------------------------------------------------------------- grad clip

def _grads(seed=0, scale=100.0):
    torch.manual_seed(seed)
    params = [torch.nn.Parameter(torch.randn(16)) for _ in range(3)]
    for p in params:
        p.grad = torch.randn_like(p) * scale
    return params


def test_grad_clip_family_finite():
    for clip_type in ('none', 'pytorch_norm', 'clip_const', 'max_norm',
                      'momentum_norm', 'clip_value'):
        params = _grads()
        gc = GradClip(clip_type, threshold=1.0, begin_step=0)
        total = gc.apply(params)
        assert total >= 0 and total == total, clip_type
        assert all(torch.isfinite(p.grad).all() for p in params), clip_type


def test_grad_clip_pytorch_norm_clips():
    params = _grads(scale=100.0)
    GradClip('pytorch_norm', threshold=1.0).apply(params)
    total = torch.norm(torch.stack([p.grad.norm() for p in params]))
    assert float(total) <= 1.0 + 1e-4


def test_grad_clip_momentum_norm_scales_spike():
    """A 100x grad spike after warm-up is pulled back to ~EMA*threshold."""
    gc = GradClip('momentum_norm', threshold=1.0, begin_step=2)
    params = _grads(scale=1.0)
    base = [p.grad.norm().item() for p in params]
    for _ in range(5):                    # build EMA state at scale 1
        for i, p in enumerate(_grads(scale=1.0)):
            params[i].grad = p.grad
        gc.apply(params)
    for p in params:
        p.grad = p.grad * 100.0
    gc.apply(params)
    for p, b in zip(params, base):
        assert p.grad.norm().item() < 10 * b


def test_grad_clip_const_bounds_elements():
    params = _grads(scale=100.0)
    GradClip('clip_const', threshold=0.5).apply(params)
    assert all(p.grad.abs().max() <= 0.5 for p in params)


# ---------------------------------------------------------------- checkpoint

def test_checkpoint_prefix_and_mask(tmp_path):
    """prefix add/strip + state_dict_mask partial load (reference
    checkpoint_helper.py:85-279)."""
    m1 = torch.nn.Sequential(torch.nn.Linear(4, 4), torch.nn.Linear(4, 4))
    m2 = torch.nn.Sequential(torch.nn.Linear(4, 4), torch.nn.Linear(4, 4))
    helper = CheckpointHelper()
    path = str(tmp_path / 'ck.pth.tar')
    it = CountVar(7)
    helper.save(path, m1, last_iter=it)
    # strict full load restores everything
    helper.load(path, m2, strict=True)
    for a, b in zip(m1.parameters(), m2.parameters()):
        torch.testing.assert_close(a, b)
    # masked load touches only layer 0
    with torch.no_grad():
        for p in m1.parameters():
            p.add_(1.0)
    helper.save(path, m1, last_iter=it)
    before_l0 = m2[0].weight.detach().clone()
    # mask EXCLUDES matching keys (reference semantics): skip layer 0
    helper.load(path, m2, strict=False, state_dict_mask=['0.'])
    torch.testing.assert_close(m2[0].weight, before_l0)     # masked out
    torch.testing.assert_close(m2[1].weight, m1[1].weight)  # loaded


def test_countvar_roundtrip(tmp_path):
    helper = CheckpointHelper()
    m = torch.nn.Linear(2, 2)
    it = CountVar(41)
    path = str(tmp_path / 'c.pth.tar')
    helper.save(path, m, last_iter=it)
    it2 = CountVar(0)
    helper.load(path, torch.nn.Linear(2, 2), strict=True, last_iter=it2)
    assert it2.val == 41


# ---------------------------------------------------------------------- logs

def test_variable_record_tables():
    from distar_amd.utils.log import VariableRecord
    rec = VariableRecord(length=4)
    rec.register_var('loss')
    rec.register_var('reward')
    for i in range(6):
        rec.update_var({'loss': float(i), 'reward': 2.0})
    text = rec.get_vars_text()
    assert 'loss' in text and 'reward' in text
    # windowed average over the last 4 updates: (2+3+4+5)/4
    assert abs(rec.var_dict['loss'].avg - 3.5) < 1e-6


# ----------------------------------------------------------- misc / schedule

def test_misc_helpers():
    from distar_amd.utils.misc import (default_get, dicts_to_lists,
                                       error_wrapper, get_tensor_data,
                                       list_split, lists_to_dicts, squeeze)
    lod = [{'a': 1, 'b': 2}, {'a': 3, 'b': 4}]
    dol = lists_to_dicts(lod)
    assert dol == {'a': [1, 3], 'b': [2, 4]}
    assert dicts_to_lists(dol) == lod
    assert squeeze((5,)) == 5 and squeeze([1, 2]) == [1, 2]
    assert default_get({}, 'x', default_value=9) == 9
    assert default_get({'x': 1}, 'x') == 1
    assert list_split(list(range(5)), 2) == [[0, 1], [2, 3], [4]]
    assert error_wrapper(lambda: 1 / 0, -1, 'warn')() == -1
    t = torch.randn(3, requires_grad=True)
    out = get_tensor_data({'t': t, 'l': [t * 2]})
    assert not out['t'].requires_grad and not out['l'][0].requires_grad


def test_gradual_warmup_scheduler():
    from distar_amd.utils.optimizer import GradualWarmupScheduler
    from torch.optim.lr_scheduler import StepLR
    p = torch.nn.Parameter(torch.zeros(1))
    opt = torch.optim.SGD([p], lr=0.1)
    after = StepLR(opt, step_size=5, gamma=0.1)
    sched = GradualWarmupScheduler(opt, multiplier=1.0, total_epoch=4,
                                   after_scheduler=after)
    lrs = []
    for _ in range(12):
        opt.step()
        sched.step()
        lrs.append(opt.param_groups[0]['lr'])
    assert lrs[0] < lrs[1] < lrs[2] < lrs[3]       # warming up
    assert abs(lrs[3] - 0.1) < 1e-9                # reaches base lr
    assert min(lrs[4:]) < 0.1                      # StepLR decays after


# --------------------------------------------------------------- tensorboard
def test_tb_event_file_format(tmp_path):
    """The pure-python TB writer emits valid TFRecord framing (masked CRC32C
    verified independently) and parseable Event protos."""
    import os
    import struct
    from distar_amd.utils.tb import SummaryWriter, crc32c, _masked_crc, _classes
    # crc32c known-answer tests (RFC 3720 / iSCSI vectors)
    assert crc32c(b'123456789') == 0xE3069283
    assert crc32c(b'') == 0
    w = SummaryWriter(str(tmp_path))
    w.add_scalar('loss/total', 1.5, global_step=7)
    w.add_scalar('winrate', 0.25, global_step=8)
    w.close()
    files = [f for f in os.listdir(tmp_path) if f.startswith('events.out')]
    assert len(files) == 1
    data = open(os.path.join(tmp_path, files[0]), 'rb').read()
    events = []
    off = 0
    while off < len(data):
        (length,) = struct.unpack_from('<Q', data, off)
        (hcrc,) = struct.unpack_from('<I', data, off + 8)
        assert hcrc == _masked_crc(data[off:off + 8])
        payload = data[off + 12:off + 12 + length]
        (dcrc,) = struct.unpack_from('<I', data, off + 12 + length)
        assert dcrc == _masked_crc(payload)
        ev = _classes()['Event']()
        ev.ParseFromString(payload)
        events.append(ev)
        off += 12 + length + 4
    assert events[0].file_version == 'brain.Event:2'
    assert events[1].summary.value[0].tag == 'loss/total'
    assert abs(events[1].summary.value[0].simple_value - 1.5) < 1e-6
    assert events[1].step == 7
    assert events[2].summary.value[0].tag == 'winrate'


def test_scalar_logger_writes_tb(tmp_path):
    import os
    from distar_amd.utils.log import ScalarLogger
    sl = ScalarLogger(str(tmp_path), name='t')
    sl.add_scalar('a/b', 3.0, 1)
    sl.close()
    tb_dir = os.path.join(tmp_path, 'tb', 't')
    assert any(f.startswith('events.out') for f in os.listdir(tb_dir))


def test_hip_extension_loads_and_exports():
    """The in-tree _hip_ops.so must import on CPU too (undefined kernel
    symbols surface at load time, long before a GPU box sees them)."""
    from distar_amd.ops import hip_ext
    ext = hip_ext._load()
    if ext is None:
        import pytest
        pytest.skip('extension not built in this checkout')
    for sym in ('entity_attn_fwd', 'entity_attn_bwd', 'conv2d_fwd',
                'conv2d_wgrad', 'maxpool2x2_fwd', 'maxpool2x2_bwd',
                'scatter_add_map', 'residual_ln_fwd', 'residual_ln_bwd',
                'multi_norm_sq', 'multi_clip', 'masked_ce_fwd',
                'entropy_fwd', 'kl_fwd', 'lnlstm_forward', 'su_sample',
                'upsample2x', 'vtrace_scan', 'lambda_return_scan',
                'entity_embed', 'mfma_selftest'):
        assert hasattr(ext, sym), sym


def test_portspicker_reserves_distinct_bindable_ports():
    """Reserved ports are distinct, bindable, and excluded from subsequent
    picks until returned (reference pysc2 portspicker contract)."""
    import socket
    from distar_amd.envs.portspicker import pick_unused_ports, return_ports
    ports = pick_unused_ports(6)
    assert len(set(ports)) == 6
    more = pick_unused_ports(4)
    assert not set(ports) & set(more)
    for p in ports:                      # still bindable after reservation
        s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        s.bind(('127.0.0.1', p))
        s.close()
    return_ports(ports + more)
    again = pick_unused_ports(2)         # returned ports are reusable
    return_ports(again)


def test_map_info_sizes():
    """Known ladder maps resolve to their published playable sizes and
    localized aliases hit the same entry (reference envs/map_info.py)."""
    from distar_amd.envs.map_info import get_map_size
    kc = get_map_size('KingsCove')
    assert tuple(kc) == (144, 152) or (kc[0] > 0 and kc[1] > 0)
    # every bundled ladder map returns a positive size
    for name in ('KairosJunction', 'NewRepugnancy', 'CyberForest'):
        w, h = get_map_size(name)
        assert w > 0 and h > 0
