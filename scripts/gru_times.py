import os
import sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from factorvae_amd.ops import get_extension
ext = get_extension()
dev = torch.device("cuda:0")

def timeit(fn, iters=50):
    fn(); torch.cuda.synchronize()
    t0=time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/iters*1e6

for N, T in [(300, 20), (3500, 60)]:
    H = 64
    gi = torch.randn(N, T, 3*H, device=dev)
    whh = torch.randn(3*H, H, device=dev) * 0.1
    whh_bf = whh.to(torch.bfloat16)
    bhh = torch.randn(3*H, device=dev) * 0.1
    hf = torch.empty(N, H, device=dev); hs = torch.empty(N, T, H, device=dev)
    hp = torch.empty(N, T, H, device=dev); g4 = torch.empty(N, T, 4*H, device=dev)
    t_f32 = timeit(lambda: ext.gru_fwd(gi, whh, bhh, hf, hs, hp, g4, N, T, H))
    t_bf = timeit(lambda: ext.gru_fwd_mfma(gi, whh_bf, bhh, hf, hs, hp, g4, N, T, H))
    dh = torch.randn(N, H, device=dev)
    dgi = torch.empty(N, T, 3*H, device=dev); dgh = torch.empty(N, T, 3*H, device=dev)
    dgib = torch.empty(N, T, 3*H, device=dev, dtype=torch.bfloat16)
    dghb = torch.empty_like(dgib)
    tb_f32 = timeit(lambda: ext.gru_bwd(dh, hp, g4, whh, dgi, dgh, N, T, H))
    tb_bf = timeit(lambda: ext.gru_bwd_mfma(dh, hp, g4, whh_bf, dgi, dgh, N, T, H, dgib, dghb))
    print(f"N={N} T={T}: fwd f32 {t_f32:.1f}us bf16 {t_bf:.1f}us | bwd f32 {tb_f32:.1f}us bf16 {tb_bf:.1f}us")
