"""GPU (MI355X) tests: HIP kernel numerics vs fp32 eager references, and the
train step end-to-end on device."""
import os
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_hip_ext_loads():
    from distar_amd.ops import hip_ext
    assert hip_ext.available(), 'in-tree _hip_ops.so must be present on GPU boxes'


def test_vtrace_scan_matches_eager():
    from distar_amd.ops import scans
    torch.manual_seed(0)
    T, B = 64, 512
    rhos = torch.rand(T, B, device='cuda')
    cs = torch.rand(T, B, device='cuda')
    r = torch.randn(T, B, device='cuda')
    v = torch.randn(T + 1, B, device='cuda')
    g = torch.rand(T, B, device='cuda')
    lam = torch.rand(T, B, device='cuda')
    out = scans.vtrace_scan(rhos, cs, r, v, g, lam)
    ref = scans._vtrace_scan_eager(rhos.cpu(), cs.cpu(), r.cpu(), v.cpu(),
                                   g.cpu(), lam.cpu())
    torch.testing.assert_close(out.cpu(), ref, rtol=1e-5, atol=1e-5)


def test_lambda_return_scan_matches_eager():
    from distar_amd.ops import scans
    torch.manual_seed(1)
    T, B = 64, 384
    r = torch.randn(T, B, device='cuda')
    g = torch.rand(T, B, device='cuda')
    v = torch.randn(T, B, device='cuda')
    lam = torch.rand(T, B, device='cuda')
    out = scans.lambda_return_scan(r, g, v, lam)
    ref = scans._lambda_return_scan_eager(r.cpu(), g.cpu(), v.cpu(), lam.cpu())
    torch.testing.assert_close(out.cpu(), ref, rtol=1e-5, atol=1e-5)


def test_sl_train_step_on_gpu():
    from distar_amd.lib.fake_data import fake_sl_batch_fast
    from distar_amd.losses import SupervisedLoss
    from distar_amd.models import Model
    from distar_amd.utils.config import Config
    from distar_amd.utils.data import to_device
    torch.manual_seed(0)
    model = Model(Config({'common': {'type': 'train'}})).cuda()
    data = to_device(fake_sl_batch_fast(batch_size=2, traj_len=4), 'cuda')
    hidden = [(torch.zeros(2, 384, device='cuda'),
               torch.zeros(2, 384, device='cuda')) for _ in range(3)]
    loss_fn = SupervisedLoss(Config({'learner': {}}))
    with torch.autocast('cuda', dtype=torch.bfloat16):
        logits, infer_action, _ = model.sl_train(
            spatial_info=data['spatial_info'], scalar_info=data['scalar_info'],
            entity_info=data['entity_info'], entity_num=data['entity_num'],
            selected_units_num=data['selected_units_num'],
            traj_lens=data['traj_lens'], hidden_state=hidden,
            action_info=data['action_info'])
        ld = loss_fn.compute_loss(logits, data['action_info'], data['action_mask'],
                                  data['selected_units_num'], data['entity_num'],
                                  infer_action)
    assert torch.isfinite(ld['total_loss'])
    ld['total_loss'].backward()
    assert all(torch.isfinite(p.grad).all() for p in model.parameters()
               if p.grad is not None)


def test_rl_train_step_on_gpu():
    from distar_amd.lib.fake_data import fake_rl_learner_data_fast
    from distar_amd.losses import ReinforcementLoss
    from distar_amd.models import Model
    from distar_amd.utils.config import Config
    from distar_amd.utils.data import to_device
    torch.manual_seed(0)
    model = Model(Config({'common': {'type': 'train'},
                          'model': {'enable_baselines':
                                    ['winloss', 'build_order', 'built_unit',
                                     'battle']}}),
                  use_value_network=True).cuda()
    data = fake_rl_learner_data_fast(2, 8, entity_num=128)
    data.pop('model_last_iter')
    data = to_device(data, 'cuda')
    loss_fn = ReinforcementLoss(Config({}), 'MP0')
    with torch.autocast('cuda', dtype=torch.bfloat16):
        out = model.rl_learner_forward(**data)
        ld = loss_fn.compute_loss(out)
    assert torch.isfinite(ld['total_loss'])
    ld['total_loss'].backward()
    assert all(torch.isfinite(p.grad).all() for p in model.parameters()
               if p.grad is not None)


def test_fused_lnlstm_matches_eager():
    """Fused HIP LN-LSTM layer vs the fp32 eager reference cell: forward
    outputs and all gradients (bf16 W_hh products => 1e-2-level tolerance)."""
    import os
    from distar_amd.models.nn.lnlstm import script_lnlstm
    torch.manual_seed(0)
    T, B, IN, H = 7, 4, 64, 384
    lstm = script_lnlstm(IN, H, 2).cuda()
    x = torch.randn(T, B, IN, device='cuda')
    states = [(torch.randn(B, H, device='cuda'), torch.randn(B, H, device='cuda'))
              for _ in range(2)]

    def run():
        xx = x.clone().requires_grad_(True)
        out, out_states = lstm(xx, [(h.clone(), c.clone()) for h, c in states])
        loss = (out.float() ** 2).mean() + sum(
            (s[0].float() ** 2).mean() + (s[1].float() ** 2).mean()
            for s in out_states)
        lstm.zero_grad()
        loss.backward()
        grads = {n: p.grad.clone() for n, p in lstm.named_parameters()}
        return out.detach().float(), grads, xx.grad.clone()

    out_hip, grads_hip, xgrad_hip = run()
    os.environ['DISTAR_AMD_DISABLE_HIP'] = '1'
    try:
        out_ref, grads_ref, xgrad_ref = run()
    finally:
        del os.environ['DISTAR_AMD_DISABLE_HIP']
    torch.testing.assert_close(out_hip, out_ref, rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(xgrad_hip, xgrad_ref, rtol=5e-2, atol=5e-2)
    for n in grads_ref:
        torch.testing.assert_close(grads_hip[n], grads_ref[n], rtol=5e-2, atol=5e-2,
                                   msg=lambda m: f'{n}: {m}')


def test_fused_lnlstm_small_hidden():
    """The selected-units pointer LSTM shape (H=32) through the same kernel.

    The eager reference here uses the SAME bf16-rounded h/W products as the
    kernel — over a 64-step recurrence, plain-fp32-vs-bf16 divergence is
    chaotic and elementwise bounds are meaningless (observed 4% mismatch),
    so the comparison has to share the rounding."""
    from distar_amd.models.nn.lnlstm import script_lnlstm
    torch.manual_seed(1)
    T, B, IN, H = 64, 128, 32, 32
    lstm = script_lnlstm(IN, H, 1).cuda()
    with torch.no_grad():
        # default randn init gives |W| ~ 1: an expansive recurrence where
        # fp32 accumulation-order noise grows ~2x per step and elementwise
        # comparison over 64 steps is meaningless.  Scale to contractive
        # dynamics (what trained weights look like).
        for cell in (l.cell for l in lstm.layers):
            cell.weight_hh.mul_(0.05)
            cell.weight_ih.mul_(0.05)
    x = torch.randn(T, B, IN, device='cuda')
    st = [(torch.zeros(B, H, device='cuda'), torch.zeros(B, H, device='cuda'))]
    out, _ = lstm(x, [(h.clone(), c.clone()) for h, c in st])

    cell = lstm.layers[0].cell
    with torch.no_grad():
        igates = cell.layernorm_i(x.reshape(T * B, -1).float()
                                  .mm(cell.weight_ih.t().float())).view(T, B, -1)
        w_bf = cell.weight_hh.bfloat16().float()
        h, c = st[0][0].clone(), st[0][1].clone()
        refs = []
        for t in range(T):
            hg = h.bfloat16().float().mm(w_bf.t())
            gates = igates[t] + cell.layernorm_h(hg)
            i, f, g, o = gates.chunk(4, 1)
            i, f, o = torch.sigmoid(i), torch.sigmoid(f), torch.sigmoid(o)
            g = torch.tanh(g)
            c = cell.layernorm_c(f * c + i * g)
            h = o * torch.tanh(c)
            refs.append(h)
        ref = torch.stack(refs)
    # early horizon: tight elementwise; full horizon: chaotic rounding
    # divergence affects a tiny tail even with the shared-rounding reference
    torch.testing.assert_close(out[:16].float(), ref[:16], rtol=2e-2, atol=1e-2)
    mismatch = ((out.float() - ref).abs() > 5e-2).float().mean()
    assert mismatch < 0.01, f'mismatch fraction {float(mismatch)}'


def test_upsample2x_matches_interpolate():
    from distar_amd.ops.upsample import upsample2x_bilinear
    torch.manual_seed(0)
    for dtype in (torch.float32, torch.bfloat16):
        x = torch.randn(3, 5, 19, 20, device='cuda', dtype=dtype, requires_grad=True)
        y = upsample2x_bilinear(x)
        ref = torch.nn.functional.interpolate(x.float(), scale_factor=2., mode='bilinear')
        torch.testing.assert_close(y.float(), ref, rtol=1e-2 if dtype == torch.bfloat16 else 1e-5,
                                   atol=1e-2 if dtype == torch.bfloat16 else 1e-5)
        g = torch.randn_like(y)
        y.backward(g)
        x2 = x.detach().float().requires_grad_(True)
        torch.nn.functional.interpolate(x2, scale_factor=2., mode='bilinear').backward(g.float())
        torch.testing.assert_close(x.grad.float(), x2.grad,
                                   rtol=1e-2 if dtype == torch.bfloat16 else 1e-4,
                                   atol=1e-2 if dtype == torch.bfloat16 else 1e-4)


def test_entity_embed_kernel_matches_eager():
    """Fused K2 entity embedding vs the 36-gather eager path."""
    import os
    from distar_amd.models import Model
    from distar_amd.lib.consts import fake_step_data
    from distar_amd.utils.config import Config
    from distar_amd.utils.data import default_collate_with_dim, to_device
    torch.manual_seed(0)
    model = Model(Config({'common': {'type': 'train'}})).cuda()
    enc = model.encoder.entity_encoder
    obs = default_collate_with_dim(
        [fake_step_data(train=False, entity_num=64, randomize=True)
         for _ in range(3)])
    ent = to_device(obs['entity_info'], 'cuda')
    ent['last_selected_units'] = torch.zeros(3, 512, dtype=torch.int8, device='cuda')
    ent['last_targeted_unit'] = torch.zeros(3, 512, dtype=torch.int8, device='cuda')
    out_hip = enc.embed_fields(ent)
    os.environ['DISTAR_AMD_DISABLE_HIP'] = '1'
    try:
        out_ref = enc.embed_fields(ent)
    finally:
        del os.environ['DISTAR_AMD_DISABLE_HIP']
    assert out_hip.shape == out_ref.shape == (3, 512, 997)
    torch.testing.assert_close(out_hip.float(), out_ref.float(),
                               rtol=1e-2, atol=1e-2)


def test_su_sample_kernel_semantics():
    """K7 sampling kernel vs the eager loop with shared uniforms: identical
    masked-logit structure at step 0, matching first picks for almost all
    rows (bf16-vs-fp32 CDF edges may flip a rare pick), and the structural
    invariants (distinct picks, end-token termination, num consistency)."""
    import os
    from distar_amd.models import Model
    from distar_amd.utils.config import Config
    torch.manual_seed(0)
    model = Model(Config({'common': {'type': 'train'}})).cuda()
    head = model.policy.selected_units_head
    B, N = 8, 80
    entity_embedding = torch.randn(B, N, 256, device='cuda')
    entity_num = torch.randint(16, N, (B,), device='cuda')
    entity_num[0] = N
    ae = torch.randn(B, 1024, device='cuda')
    su_mask = torch.ones(B, dtype=torch.bool, device='cuda')
    su_mask[3] = False
    uniforms = torch.rand(B, 64, device='cuda')
    with torch.no_grad():
        key, mask, key_emb = head._get_key_mask(entity_embedding, entity_num)
        logits_h, results_h, ae_h, num_h, extra_h = head._query_sample_hip(
            key, entity_num, ae, mask, key_emb, su_mask, uniforms=uniforms)
        os.environ['DISTAR_AMD_DISABLE_HIP'] = '1'
        try:
            logits_e, results_e, ae_e, num_e, extra_e = head._query_sample(
                key, entity_num, ae, mask, key_emb, su_mask, uniforms=uniforms)
        finally:
            del os.environ['DISTAR_AMD_DISABLE_HIP']
    assert int(num_h[3]) == 0 and int(num_e[3]) == 0
    live = su_mask.nonzero().squeeze(1)
    # step-0 logits match closely where unmasked
    l0h, l0e = logits_h[live, 0], logits_e[live, 0]
    sel = l0e > -1e8
    torch.testing.assert_close(l0h[sel], l0e[sel], rtol=5e-2, atol=5e-2)
    assert (l0h[~sel] < -1e8).all()
    # Exact agreement on every CDF-stable pick: a pick can only legally
    # diverge when the shared uniform lands within eps of an (eager) CDF
    # boundary — there bf16/fp32 rounding differences between the kernel
    # and eager softmax flip the inverse-CDF bin.  Away from boundaries
    # the kernel MUST reproduce eager exactly, step by step, while the
    # prefix agrees.  (replaces the 0.7 first-pick gate.)
    eps = 0.05
    probs_e = torch.softmax(logits_e.float(), dim=-1)
    stable_total = stable_agree = 0
    for b in live.tolist():
        steps = min(int(num_e[b]), int(num_h[b]))
        for s in range(steps):
            if s > 0 and not torch.equal(results_h[b, :s], results_e[b, :s]):
                break
            cdf = probs_e[b, s].cumsum(0)
            u = float(uniforms[b, s])
            dist = float((cdf - u).abs().min())
            if dist > eps:
                stable_total += 1
                stable_agree += int(results_h[b, s] == results_e[b, s])
    assert stable_total >= 10, 'stability check is vacuous'
    assert stable_agree == stable_total, \
        f'{stable_agree}/{stable_total} stable picks agree'
    # invariants per live row
    for b in live.tolist():
        n = int(num_h[b])
        seq = results_h[b, :n].tolist()
        body = [s for s in seq if s != int(entity_num[b])]
        assert len(body) == len(set(body))
        if n < 64:
            assert seq[-1] == int(entity_num[b])
    # determinism: same uniforms -> identical kernel outputs
    with torch.no_grad():
        logits2, results2, _, num2, _ = head._query_sample_hip(
            key, entity_num, ae, mask, key_emb, su_mask, uniforms=uniforms)
    assert torch.equal(results2, results_h) and torch.equal(num2, num_h)
    # rows whose full sequences agree must also agree on final ae
    full = [b for b in live.tolist()
            if int(num_h[b]) == int(num_e[b]) and
            torch.equal(results_h[b, :int(num_h[b])], results_e[b, :int(num_e[b])])]
    assert full, 'no fully-agreeing rows to compare'
    torch.testing.assert_close(ae_h[full], ae_e[full], rtol=5e-2, atol=5e-2)


@pytest.mark.gpu
def test_fused_masked_ce_matches_eager():
    """K13 fused masked CE vs F.cross_entropy fp32 (forward + backward)."""
    import distar_amd.ops.ce_loss as ce
    torch.manual_seed(0)
    N, C = 64, 24320
    logits = torch.randn(N, C, device='cuda') * 3
    labels = torch.randint(0, C, (N,), device='cuda')
    mask = (torch.rand(N, device='cuda') > 0.3).float()
    ref_l = logits.detach().clone().requires_grad_(True)
    ref = torch.nn.functional.cross_entropy(ref_l, labels,
                                            reduction='none') * mask
    ref.sum().backward()
    os.environ['DISTAR_AMD_FUSED_CE'] = '1'   # explicit (also the default)
    try:
        fused_l = logits.detach().clone().requires_grad_(True)
        out = ce.masked_cross_entropy(fused_l, labels, mask)
        torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-5)
        out.sum().backward()
        torch.testing.assert_close(fused_l.grad, ref_l.grad,
                                   rtol=1e-4, atol=1e-6)
    finally:
        os.environ.pop('DISTAR_AMD_FUSED_CE', None)


@pytest.mark.gpu
def test_fused_rowwise_entropy_kl_match_eager():
    """K13 fused entropy/KL vs eager fp32 (forward + backward)."""
    from distar_amd.ops.rl_rowwise import rowwise_entropy, rowwise_kl
    torch.manual_seed(0)
    N, C = 64, 24320
    t = torch.randn(N, C, device='cuda') * 2
    s_ref = (torch.randn(N, C, device='cuda') * 2).requires_grad_(True)
    os.environ['DISTAR_AMD_FUSED_RL_ROWWISE'] = '0'
    ent_ref = rowwise_entropy(s_ref)            # eager reference
    kl_ref = rowwise_kl(t, s_ref)
    (ent_ref.sum() + kl_ref.sum()).backward()
    os.environ['DISTAR_AMD_FUSED_RL_ROWWISE'] = '1'
    try:
        s = s_ref.detach().clone().requires_grad_(True)
        ent = rowwise_entropy(s)
        kl = rowwise_kl(t, s)
        torch.testing.assert_close(ent, ent_ref, rtol=1e-5, atol=1e-5)
        torch.testing.assert_close(kl, kl_ref, rtol=1e-5, atol=1e-5)
        (ent.sum() + kl.sum()).backward()
        torch.testing.assert_close(s.grad, s_ref.grad, rtol=1e-4, atol=1e-6)
    finally:
        os.environ.pop('DISTAR_AMD_FUSED_RL_ROWWISE', None)


# ------------------------------------------------------ K1 entity attention
def test_mfma_selftest_layout():
    """One 16x16x32 bf16 MFMA vs torch matmul — isolates fragment-layout
    mistakes from attention logic (asymmetric random inputs per the guide's
    transpose-detection rule)."""
    from distar_amd.ops import hip_ext
    ops = hip_ext.require()
    torch.manual_seed(0)
    A = torch.randn(16, 32, device='cuda').bfloat16()
    B = torch.randn(16, 32, device='cuda').bfloat16()
    D = ops.mfma_selftest(A, B)
    ref = A.float() @ B.float().T
    torch.testing.assert_close(D, ref, rtol=2e-2, atol=2e-2)


def _eager_attn_fp32(qkv, entity_num, H, scale):
    """fp32 reference with the reference's additive -1e9 convention."""
    B, N, F3 = qkv.shape
    D = F3 // 3 // H
    q, k, v = qkv.float().split(F3 // 3, dim=-1)
    q = q.view(B, N, H, D).permute(0, 2, 1, 3)
    k = k.view(B, N, H, D).permute(0, 2, 1, 3)
    v = v.view(B, N, H, D).permute(0, 2, 1, 3)
    s = torch.matmul(q, k.transpose(-2, -1)) * scale
    if entity_num is not None:
        key = torch.arange(N, device=qkv.device).view(1, 1, 1, N)
        s = s + torch.where(key < entity_num.view(B, 1, 1, 1), 0.0, -1e9)
    p = torch.softmax(s, dim=-1)
    o = torch.matmul(p, v)
    return o.permute(0, 2, 1, 3).reshape(B, N, H * D)


@pytest.mark.parametrize('N,nums', [
    (512, [512, 300, 64, 1, 17]),
    (256, [256, 250, 100, 3, 256]),
    (160, [160, 96, 33, 160, 2]),      # N not a multiple of 64
])
def test_entity_attn_fwd_matches_eager(N, nums):
    from distar_amd.ops.entity_attn import entity_attention
    torch.manual_seed(0)
    B, H, D = 5, 2, 128
    qkv = (torch.randn(B, N, 3 * H * D, device='cuda') * 0.5).bfloat16()
    en = torch.tensor(nums[:B], device='cuda', dtype=torch.int32)
    out = entity_attention(qkv, en, H, 1.0 / D ** 0.5)
    ref = _eager_attn_fp32(qkv, en, H, 1.0 / D ** 0.5)
    torch.testing.assert_close(out.float(), ref, rtol=3e-2, atol=3e-2)
    # unmasked path too
    out2 = entity_attention(qkv, None, H, 1.0 / D ** 0.5)
    ref2 = _eager_attn_fp32(qkv, None, H, 1.0 / D ** 0.5)
    torch.testing.assert_close(out2.float(), ref2, rtol=3e-2, atol=3e-2)


def test_entity_attn_bwd_matches_eager():
    from distar_amd.ops.entity_attn import entity_attention
    torch.manual_seed(1)
    B, N, H, D = 4, 512, 2, 128
    scale = 1.0 / D ** 0.5
    qkv0 = (torch.randn(B, N, 3 * H * D, device='cuda') * 0.5).bfloat16()
    en = torch.tensor([512, 317, 65, 1], device='cuda', dtype=torch.int32)
    dout = (torch.randn(B, N, H * D, device='cuda') * 0.5).bfloat16()

    qkv_h = qkv0.detach().clone().requires_grad_(True)
    out_h = entity_attention(qkv_h, en, H, scale)
    out_h.backward(dout)

    qkv_e = qkv0.detach().clone().float().requires_grad_(True)
    out_e = _eager_attn_fp32(qkv_e, en, H, scale)
    out_e.backward(dout.float())

    torch.testing.assert_close(out_h.float(), out_e.detach(),
                               rtol=3e-2, atol=3e-2)
    # bf16 recompute backward vs fp32 eager: absolute tolerance scaled to
    # the gradient magnitude
    g_h, g_e = qkv_h.grad.float(), qkv_e.grad
    scale_tol = g_e.abs().max().item()
    torch.testing.assert_close(g_h, g_e, rtol=5e-2, atol=0.05 * scale_tol)


def test_entity_transformer_hip_vs_eager_end_to_end():
    """The full 3-layer entity Transformer module: HIP attention path vs
    DISTAR_AMD_DISABLE_HIP=1 eager, same bf16 weights."""
    from distar_amd.models.nn.blocks import sequence_mask
    from distar_amd.models.nn.transformer import Transformer
    torch.manual_seed(2)
    tr = Transformer(input_dim=997, head_dim=128, hidden_dim=1024,
                     output_dim=256, head_num=2, mlp_num=2, layer_num=3,
                     ln_type='post').cuda()
    x = torch.randn(3, 512, 997, device='cuda')
    en = torch.tensor([512, 200, 7], device='cuda')
    mask = sequence_mask(en, max_len=512)
    with torch.autocast('cuda', dtype=torch.bfloat16):
        out_hip = tr(x, mask=mask)
    os.environ['DISTAR_AMD_DISABLE_HIP'] = '1'
    try:
        with torch.autocast('cuda', dtype=torch.bfloat16):
            out_eager = tr(x, mask=mask)
    finally:
        os.environ.pop('DISTAR_AMD_DISABLE_HIP', None)
    # valid rows must agree; padded rows are don't-care downstream (the
    # encoder masks them before reduction)
    m = mask.unsqueeze(-1)
    torch.testing.assert_close((out_hip * m).float(), (out_eager * m).float(),
                               rtol=5e-2, atol=5e-2)


def test_scatter_add_map_matches_eager():
    """K3 HIP NCHW scatter-add (packed bf16 atomics) vs the eager index_add
    formulation, forward + backward."""
    from distar_amd.ops.scatter import scatter_connection
    torch.manual_seed(3)
    B, N, C, H, W = 6, 512, 32, 152, 160
    emb0 = (torch.randn(B, N, C, device='cuda') * 0.5).bfloat16()
    loc = torch.stack([torch.randint(0, W, (B, N), device='cuda'),
                       torch.randint(0, H, (B, N), device='cuda')], dim=-1)
    # force some collisions
    loc[:, 1] = loc[:, 0]
    emb_h = emb0.detach().clone().requires_grad_(True)
    out_h = scatter_connection((B, H, W), emb_h, loc, C, 'add')
    dout = (torch.randn_like(out_h) * 0.5).bfloat16()
    out_h.backward(dout)
    os.environ['DISTAR_AMD_DISABLE_HIP'] = '1'
    try:
        emb_e = emb0.detach().clone().requires_grad_(True)
        out_e = scatter_connection((B, H, W), emb_e, loc, C, 'add')
        out_e.backward(dout)
    finally:
        os.environ.pop('DISTAR_AMD_DISABLE_HIP', None)
    torch.testing.assert_close(out_h.float(), out_e.float(),
                               rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(emb_h.grad.float(), emb_e.grad.float(),
                               rtol=2e-2, atol=2e-2)


def test_fused_residual_ln_matches_eager():
    """Fused residual-add + LayerNorm vs fp32 eager (fwd + bwd, incl. the
    LN parameter grads accumulated via atomics)."""
    from distar_amd.ops.residual_ln import fused_residual_ln
    torch.manual_seed(4)
    R, C = 3000, 256
    ln = torch.nn.LayerNorm(C).cuda()
    x0 = (torch.randn(R, C, device='cuda') * 0.7).bfloat16()
    a0 = (torch.randn(R, C, device='cuda') * 0.7).bfloat16()
    dout = (torch.randn(R, C, device='cuda') * 0.5).bfloat16()

    x_h = x0.detach().clone().requires_grad_(True)
    a_h = a0.detach().clone().requires_grad_(True)
    y_h = fused_residual_ln(x_h, a_h, ln)
    y_h.backward(dout)
    gw_h, gb_h = ln.weight.grad.clone(), ln.bias.grad.clone()
    ln.weight.grad = None
    ln.bias.grad = None

    x_e = x0.detach().clone().float().requires_grad_(True)
    a_e = a0.detach().clone().float().requires_grad_(True)
    y_e = torch.nn.functional.layer_norm(x_e + a_e, (C,), ln.weight, ln.bias,
                                         ln.eps)
    y_e.backward(dout.float())

    torch.testing.assert_close(y_h.float(), y_e.detach(), rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(x_h.grad.float(), x_e.grad, rtol=5e-2,
                               atol=5e-2)
    torch.testing.assert_close(a_h.grad.float(), a_e.grad, rtol=5e-2,
                               atol=5e-2)
    # parameter grads sum over 3000 rows: scale tolerance to magnitude
    tol = ln.weight.grad.abs().max().item() if False else None
    torch.testing.assert_close(gw_h, ln.weight.grad, rtol=2e-2,
                               atol=0.02 * float(ln.weight.grad.abs().max()))
    torch.testing.assert_close(gb_h, ln.bias.grad, rtol=2e-2,
                               atol=0.02 * float(ln.bias.grad.abs().max()))


# --------------------------------------------------------------- K4 conv
@pytest.mark.parametrize('cin,cout,hw,kh', [
    (56, 32, (152, 160), 1),     # spatial project
    (32, 64, (76, 80), 3),       # downsample convs
    (128, 128, (19, 20), 3),     # ResBlocks
    (132, 128, (19, 20), 1),     # location head 1x1
    (32, 1, (76, 80), 3),        # stencil path, ragged 16-col tail
    (32, 1, (152, 160), 3),      # stencil path at the real location shape
    (24, 2, (76, 80), 3),        # generic small-Cout fallback kernels
    (48, 32, (19, 20), 3),       # windowed wgrad, ragged k-tiles (K=432)
    (16, 16, (16, 16), 3),       # windowed wgrad, 18x18 padded window
])
def test_conv2d_hip_matches_eager(cin, cout, hw, kh):
    """K4 MFMA implicit-GEMM conv vs fp32 F.conv2d (fwd + both bwds)."""
    from distar_amd.ops.conv2d import Conv2dHIP
    torch.manual_seed(5)
    H, W = hw
    B = 3
    conv = Conv2dHIP(cin, cout, kh, 1, padding=kh // 2).cuda()
    x0 = (torch.randn(B, cin, H, W, device='cuda') * 0.5).bfloat16()
    dout = (torch.randn(B, cout, H, W, device='cuda') * 0.5).bfloat16()

    x_h = x0.detach().clone().requires_grad_(True)
    out_h = conv(x_h)
    out_h.backward(dout)
    gw_h = conv.weight.grad.clone()
    gb_h = conv.bias.grad.clone()
    conv.weight.grad = None
    conv.bias.grad = None

    x_e = x0.detach().clone().float().requires_grad_(True)
    out_e = torch.nn.functional.conv2d(x_e, conv.weight, conv.bias,
                                       padding=kh // 2)
    out_e.backward(dout.float())

    torch.testing.assert_close(out_h.float(), out_e.detach(),
                               rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(x_h.grad.float(), x_e.grad.float(),
                               rtol=5e-2, atol=5e-2)
    sw = float(conv.weight.grad.abs().max())
    torch.testing.assert_close(gw_h, conv.weight.grad, rtol=3e-2,
                               atol=0.03 * sw)
    sb = float(conv.bias.grad.abs().max())
    torch.testing.assert_close(gb_h, conv.bias.grad, rtol=3e-2,
                               atol=0.03 * sb)


def test_maxpool2x2_hip_matches_eager():
    from distar_amd.ops.conv2d import max_pool2x2
    torch.manual_seed(6)
    x0 = (torch.randn(4, 32, 76, 80, device='cuda')).bfloat16()
    x_h = x0.detach().clone().requires_grad_(True)
    out_h = max_pool2x2(x_h)
    dout = torch.randn_like(out_h).bfloat16()
    out_h.backward(dout)
    x_e = x0.detach().clone().requires_grad_(True)
    os.environ['DISTAR_AMD_CONV'] = '0'
    try:
        out_e = max_pool2x2(x_e)
        out_e.backward(dout)
    finally:
        os.environ.pop('DISTAR_AMD_CONV', None)
    torch.testing.assert_close(out_h.float(), out_e.float())
    torch.testing.assert_close(x_h.grad.float(), x_e.grad.float())


def test_multi_tensor_norm_clip_matches_eager():
    """K14 multi-tensor global-norm + clip vs clip_grad_norm_."""
    from distar_amd.ops import hip_ext
    ops = hip_ext.require()
    torch.manual_seed(7)
    grads = [torch.randn(n, device='cuda') * 3
             for n in (17, 1024, 100003, 4096)]
    ref = [g.clone() for g in grads]
    norm_sq = ops.multi_norm_sq(grads)
    expected_norm = torch.norm(torch.cat([g.view(-1) for g in ref]))
    torch.testing.assert_close(norm_sq.sqrt(), expected_norm,
                               rtol=1e-5, atol=1e-5)
    ops.multi_clip(grads, norm_sq, 1.0, 1e-6)
    scale = min(1.0, 1.0 / (float(expected_norm) + 1e-6))
    for g, r in zip(grads, ref):
        torch.testing.assert_close(g, r * scale, rtol=1e-5, atol=1e-6)


@pytest.mark.gpu
def test_conv2d_fused_relu_matches_eager():
    """conv2d_block folds ReLU into the conv epilogue (fwd) and applies
    the mask via threshold_backward on the saved output (bwd)."""
    import torch.nn.functional as F
    from distar_amd.models.nn.blocks import conv2d_block
    torch.manual_seed(11)
    blk = conv2d_block(32, 64, 3, 1, 1, activation='relu').cuda()
    conv = blk[0]
    assert getattr(conv, 'fuse_relu', False) and len(blk) == 1
    x0 = (torch.randn(2, 32, 76, 80, device='cuda') * 0.5).bfloat16()
    dout = (torch.randn(2, 64, 76, 80, device='cuda') * 0.5).bfloat16()

    x_h = x0.detach().clone().requires_grad_(True)
    out_h = blk(x_h)
    out_h.backward(dout)
    gw, gb = conv.weight.grad.clone(), conv.bias.grad.clone()
    conv.weight.grad = None
    conv.bias.grad = None

    out_e = F.relu(F.conv2d(x0.float(), conv.weight, conv.bias, padding=1))
    torch.testing.assert_close(out_h.float(), out_e, rtol=3e-2, atol=3e-2)
    # grads: elements with pre-relu output ~0 get their mask flipped by
    # bf16-vs-fp32 accumulation-order noise, so the reference must use the
    # KERNEL's mask (otherwise ~1% of x.grad differs by the full |dy|)
    dy_m = dout.float() * (out_h.detach().float() > 0)
    ref_dx = torch.nn.grad.conv2d_input(x0.shape, conv.weight, dy_m,
                                        padding=1)
    ref_dw = torch.nn.grad.conv2d_weight(x0.float(), conv.weight.shape,
                                         dy_m, padding=1)
    ref_db = dy_m.sum(dim=(0, 2, 3))
    torch.testing.assert_close(x_h.grad.float(), ref_dx, rtol=5e-2,
                               atol=5e-2)
    torch.testing.assert_close(gw, ref_dw, rtol=3e-2,
                               atol=0.03 * float(ref_dw.abs().max()))
    torch.testing.assert_close(gb, ref_db, rtol=3e-2,
                               atol=0.03 * float(ref_db.abs().max()))


@pytest.mark.gpu
def test_fused_linear_relu_matches_eager():
    """fc_block's [Linear, ReLU] runs through the hipBLASLt relu epilogue
    with a hand-supplied backward (threshold mask on the saved output)."""
    from distar_amd.models.nn.blocks import fc_block
    from distar_amd.ops.linear_relu import FusedLinearReLU
    torch.manual_seed(3)
    blk = fc_block(256, 512, activation='relu').cuda()
    assert isinstance(blk[0], FusedLinearReLU)
    lin = blk[0]
    x0 = (torch.randn(6, 33, 256, device='cuda') * 0.5).bfloat16()
    dout = (torch.randn(6, 33, 512, device='cuda') * 0.5).bfloat16()

    x_h = x0.detach().clone().requires_grad_(True)
    out_h = blk(x_h)
    out_h.backward(dout)
    gw, gb = lin.weight.grad.clone(), lin.bias.grad.clone()

    out_e = torch.relu(torch.nn.functional.linear(
        x0.float(), lin.weight, lin.bias))
    torch.testing.assert_close(out_h.float(), out_e, rtol=3e-2, atol=3e-2)
    dy_m = dout.float() * (out_h.detach().float() > 0)   # kernel's mask
    ref_dx = dy_m @ lin.weight
    ref_dw = dy_m.reshape(-1, 512).t() @ x0.float().reshape(-1, 256)
    ref_db = dy_m.sum(dim=(0, 1))
    torch.testing.assert_close(x_h.grad.float(), ref_dx, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(gw, ref_dw, rtol=3e-2,
                               atol=0.03 * float(ref_dw.abs().max()))
    torch.testing.assert_close(gb, ref_db, rtol=3e-2,
                               atol=0.03 * float(ref_db.abs().max()))


@pytest.mark.gpu
def test_film_location_head_autocast():
    """FiLM-conditioned LocationHead (the alternative league pipeline's
    location path) under cuda autocast vs the CPU fp32 reference —
    closes the SURVEY/roadmap gap of the FiLM path having no GPU test."""
    import copy
    from distar_amd.models.alphastar.heads import LocationHead
    from distar_amd.utils.config import Config, read_config
    import os as _os
    cfg = read_config(_os.path.join(
        _os.path.dirname(__file__), '..', 'distar_amd', 'models',
        'alphastar', 'actor_critic_default_config.yaml'))
    whole = Config(copy.deepcopy(dict(cfg)))
    whole.model.policy.head.location_head.film = True
    whole.model.policy.head.location_head.gate = False
    torch.manual_seed(7)
    head_cpu = LocationHead(whole).float()
    head_gpu = copy.deepcopy(head_cpu).cuda()
    B = 3
    emb = torch.randn(B, 1024)
    skips = [torch.randn(B, 128, 19, 20) * 0.3 for _ in range(7)]
    loc = torch.randint(0, 152 * 160, (B,))

    logits_cpu, _ = head_cpu(emb, skips, loc)
    with torch.autocast('cuda', dtype=torch.bfloat16):
        logits_gpu, _ = head_gpu(emb.cuda(), [s.cuda() for s in skips],
                                 loc.cuda())
    assert torch.isfinite(logits_gpu).all()
    # bf16 through 4 FiLM res stages + upsample chain: loose elementwise
    torch.testing.assert_close(logits_gpu.float().cpu(), logits_cpu,
                               rtol=8e-2, atol=8e-2)
    logits_gpu.float().sum().backward()
    assert all(torch.isfinite(p.grad).all() for p in head_gpu.parameters()
               if p.grad is not None)
