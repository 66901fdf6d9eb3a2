import os
import sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from factorvae_amd.data.sampler import init_data_loader
from factorvae_amd.data.synthetic import make_synthetic_frame
from factorvae_amd.models.modules import build_factorvae
from factorvae_amd.engine.fused import FusedTrainer

DEV = torch.device("cuda:0")
df = make_synthetic_frame(n_days=120, n_stocks=300, seed=4)
model = build_factorvae(num_latent=158, hidden_size=64, num_portfolio=128, num_factor=20).to(DEV)
loader = init_data_loader(df, step_len=20, shuffle=False, start=None, end=None)

t0=time.perf_counter()
batches = [(x[:, :, :-1].contiguous()) for x,_ in loader]
t1=time.perf_counter()
print(f"loader+slice: {len(batches)} days in {t1-t0:.2f}s = {len(batches)/(t1-t0):.0f}/s")

trainer = FusedTrainer(model, lr=0.0, t_max=1, device=DEV, train=False)
x0 = batches[0].to(DEV)
trainer.predict(x0); torch.cuda.synchronize()
t0=time.perf_counter()
for b in batches:
    xb = b.to(DEV)
    p = trainer.predict(xb)
torch.cuda.synchronize()
t1=time.perf_counter()
print(f"h2d+predict: {len(batches)/(t1-t0):.0f}/s")
t0=time.perf_counter()
for b in batches:
    p = trainer.predict(x0)
torch.cuda.synchronize()
t1=time.perf_counter()
print(f"predict-only: {len(batches)/(t1-t0):.0f}/s")
t0=time.perf_counter()
for b in batches:
    xb = b.to(DEV)
    p = trainer.predict(xb).detach().cpu()
t1=time.perf_counter()
print(f"full w/ d2h: {len(batches)/(t1-t0):.0f}/s")
m2 = model
m2.eval()
with torch.no_grad():
    m2.prediction(x0); torch.cuda.synchronize()
    t0=time.perf_counter()
    for b in batches[:30]:
        p = m2.prediction(b.to(DEV))
    torch.cuda.synchronize()
    t1=time.perf_counter()
print(f"eager prediction: {30/(t1-t0):.0f}/s")
