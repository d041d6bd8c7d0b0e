import sys, time
sys.path.insert(0, '.')
import torch
torch.set_num_threads(1)
from pdrl_amd.agents import Worker
from pdrl_amd.transport import Endpoint
from pdrl_amd.utils import load_params
import main as main_mod

p = load_params(); p.algo = "IMPALA"; p.env = "CartPole-v1"
main_mod.probe_env_spaces(p)
model = main_mod.build_model(p)
sub = Endpoint(bind=("127.0.0.1", 0))
w = Worker(model, 0, "127.0.0.1", sub.bound_port, "127.0.0.1", 1, p, seed=0)
t0 = time.perf_counter()
w.collect(max_episodes=40)
dt = time.perf_counter() - t0
n = 0
while sub.recv(timeout=0.2) is not None: n += 1
print(f"isolated worker: {n} steps in {dt:.2f}s -> {n/dt:.0f} steps/s")
obs = torch.zeros(1, 4); hx = torch.zeros(1, 64); cx = torch.zeros(1, 64)
t0 = time.perf_counter()
for _ in range(500):
    w.model.act(obs, (hx, cx))
print(f"model.act: {(time.perf_counter()-t0)/500*1e6:.0f} us")
