import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

def t(fn, it=3):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(it): r = fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / it

for n in (846, 2900):
    a = torch.rand(n, n, dtype=torch.float64, device="cuda") + n * torch.eye(n, dtype=torch.float64, device="cuda")
    eye = torch.eye(n, dtype=torch.float64, device="cuda")
    print(f"n={n}: inv {t(lambda: torch.linalg.inv(a))*1e3:7.1f} ms   "
          f"solve(a,I) {t(lambda: torch.linalg.solve(a, eye))*1e3:7.1f} ms   "
          f"lu_factor {t(lambda: torch.linalg.lu_factor(a))*1e3:7.1f} ms   "
          f"host_inv {t(lambda: torch.from_numpy(__import__('numpy').linalg.inv(a.cpu().numpy())).cuda())*1e3:7.1f} ms", flush=True)
