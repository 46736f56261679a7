import sys, time, torch
sys.path.insert(0, '.')
from distar_amd.models.nn.lnlstm import script_lnlstm

def run(tag, T, B, IN, H, layers):
    torch.manual_seed(0)
    lstm = script_lnlstm(IN, H, layers).cuda()
    x = torch.randn(T, B, IN, device='cuda', requires_grad=True)
    st = [(torch.zeros(B, H, device='cuda'), torch.zeros(B, H, device='cuda')) for _ in range(layers)]
    torch.cuda.synchronize(); t0 = time.time()
    out, _ = lstm(x, st)
    torch.cuda.synchronize(); t1 = time.time()
    loss = out.float().square().mean()
    loss.backward()
    torch.cuda.synchronize(); t2 = time.time()
    print(f'{tag}: fwd {1000*(t1-t0):.1f} ms  bwd {1000*(t2-t1):.1f} ms', flush=True)

print('start', flush=True)
run('core  T=64 B=32', 64, 32, 1536, 384, 3)
run('core2 T=64 B=32', 64, 32, 1536, 384, 3)
run('su    T=64 B=2048', 64, 2048, 32, 32, 1)
run('su2   T=64 B=2048', 64, 2048, 32, 32, 1)
import os
os.environ['DISTAR_AMD_DISABLE_HIP'] = '1'
run('core eager', 64, 32, 1536, 384, 3)
run('su eager', 64, 2048, 32, 32, 1)
