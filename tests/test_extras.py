"""Auxiliary components: soft-argmax, extra losses, extended Adam, locks."""
import torch

from distar_amd.losses.extra import LabelSmoothCELoss, MultiLogitsLoss, SoftFocalLoss
from distar_amd.models.nn.extras import SoftArgmax
from distar_amd.utils.locks import LockContext, LockContextType
from distar_amd.utils.optimizer import Adam


def test_soft_argmax_peaks():
    x = torch.full((2, 1, 8, 10), -20.)
    x[0, 0, 3, 7] = 20.
    x[1, 0, 5, 2] = 20.
    out = SoftArgmax()(x)
    torch.testing.assert_close(out, torch.tensor([[3., 7.], [5., 2.]]),
                               rtol=1e-3, atol=1e-3)


def test_extra_losses():
    torch.manual_seed(0)
    logits = torch.randn(6, 10, requires_grad=True)
    labels = torch.arange(6)
    for loss_fn in (LabelSmoothCELoss(0.1), SoftFocalLoss()):
        loss = loss_fn(logits, labels)
        assert torch.isfinite(loss)
        loss.backward(retain_graph=True)
    ml = MultiLogitsLoss()(torch.randn(4, 9), torch.tensor([1, 3, 5, 7]))
    assert torch.isfinite(ml)


def test_extended_adam_with_clip():
    torch.manual_seed(0)
    p = torch.nn.Parameter(torch.randn(10))
    opt = Adam([p], lr=0.1, grad_clip_type='clip_const', clip_value=0.01)
    before = p.detach().clone()
    (p * 100).sum().backward()
    opt.step()
    # const-clip bounds the per-element grad to 0.01 -> step ~ lr bounded
    assert (p.detach() - before).abs().max() < 0.2


def test_lock_context():
    with LockContext(LockContextType.THREAD_LOCK):
        pass
    with LockContext(LockContextType.PROCESS_LOCK):
        pass
