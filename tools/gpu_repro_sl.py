import faulthandler, sys, time, os
faulthandler.dump_traceback_later(int(os.environ.get('WD', '150')), exit=True)
sys.path.insert(0, '.')
import torch
from distar_amd.lib.fake_data import fake_sl_batch_fast
from distar_amd.losses import SupervisedLoss
from distar_amd.models import Model
from distar_amd.utils.config import Config
from distar_amd.utils.data import to_device

torch.manual_seed(0)
dev = 'cuda'
model = Model(Config({'common': {'type': 'train'}})).to(dev)
B, T = int(os.environ.get('B', 32)), int(os.environ.get('T', 64))
data = to_device(fake_sl_batch_fast(B, T), dev)
hidden = [(torch.zeros(B, 384, device=dev), torch.zeros(B, 384, device=dev)) for _ in range(3)]
loss_fn = SupervisedLoss(Config({'learner': {}}))
amp = os.environ.get('AMP', '1') == '1'
def sync(tag):
    torch.cuda.synchronize(); print(tag, time.time() - t0, flush=True)
t0 = time.time()
with torch.autocast('cuda', dtype=torch.bfloat16, enabled=amp):
    out = model.encoder(data['spatial_info'], data['entity_info'], data['scalar_info'], data['entity_num'])
    sync('encoder')
    lstm_input, scalar_context, baseline_feature, entity_embeddings, map_skip = out
    li = lstm_input.view(-1, lstm_input.shape[0] // B, lstm_input.shape[-1]).permute(1, 0, 2)
    lstm_out, out_state = model.core_lstm(li, hidden)
    sync('core_lstm fwd')
    lstm_out = lstm_out.permute(1, 0, 2).contiguous().view(-1, lstm_out.shape[-1])
    action_info, sun, logits = model.policy.train_forward(
        lstm_out, entity_embeddings, map_skip, scalar_context, data['entity_num'],
        data['action_info'], data['selected_units_num'])
    sync('policy fwd')
    ld = loss_fn.compute_loss(logits, data['action_info'], data['action_mask'],
                              data['selected_units_num'], data['entity_num'], action_info)
    sync('loss')
ld['total_loss'].backward()
sync('backward')
print('DONE', flush=True)
