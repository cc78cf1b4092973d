import os, sys, time
sys.path.insert(0, os.getcwd())
import numpy as np
from optuna_amd import _hip
core = _hip.get()
print("core:", core is not None and core.available())
from optuna_amd._hypervolume import hssp
rng = np.random.RandomState(0)
n, k = 2500, 600
vals = rng.rand(n, 3)
idx = np.arange(n)
ref = np.array([1.2, 1.2, 1.2])
t0 = time.perf_counter()
out = hssp._solve_hssp_3d_device(vals, idx, k, ref)
t1 = time.perf_counter()
print("device greedy:", None if out is None else len(out), f"{(t1-t0)*1e3:.1f} ms")
t0 = time.perf_counter()
out2 = hssp._solve_hssp(vals, idx, k, ref)
t1 = time.perf_counter()
print("_solve_hssp routed:", f"{(t1-t0)*1e3:.1f} ms", "equal:", out is not None and set(out)==set(out2))
