import os, sys
sys.path.insert(0, os.getcwd())
import numpy as np, warnings, math, time
warnings.simplefilter("ignore")
import optuna_amd
from optuna_amd._hypervolume import hssp
optuna_amd.logging.set_verbosity(optuna_amd.logging.WARNING)

orig = hssp._solve_hssp_3d_device
def spy(vals, idx, subset, ref):
    t0 = time.perf_counter()
    out = orig(vals, idx, subset, ref)
    print(f"[hssp] n={len(vals)} M={vals.shape[1]} subset={subset} ref={ref} -> "
          f"{'DEVICE' if out is not None else 'None'} {(time.perf_counter()-t0)*1e3:.1f}ms", flush=True)
    return out
hssp._solve_hssp_3d_device = spy
orig_solve = hssp._solve_hssp
def spy2(vals, idx, subset, ref):
    t0 = time.perf_counter()
    out = orig_solve(vals, idx, subset, ref)
    print(f"[solve] n={len(vals)} subset={subset} took {(time.perf_counter()-t0)*1e3:.1f}ms", flush=True)
    return out
hssp._solve_hssp = spy2
# re-import sites bind at call time via module attr? _history imports the symbol INSIDE the function:
# "from optuna_amd._hypervolume.hssp import _solve_hssp" per call -> picks up spy2 ✓

rng = np.random.RandomState(0)
n_hist = 6000
names = [f"x{i}" for i in range(10)]
dists = {n: optuna_amd.distributions.FloatDistribution(0.0, 1.0) for n in names}
study = optuna_amd.create_study(directions=["minimize"]*3,
                                sampler=optuna_amd.samplers.TPESampler(seed=0, n_startup_trials=10))
pm = rng.uniform(0, 1, size=(n_hist, 10))
def dtlz2(row):
    g = float(np.sum((row[2:] - 0.5) ** 2))
    return [(1+g)*math.cos(row[0]*math.pi/2)*math.cos(row[1]*math.pi/2),
            (1+g)*math.cos(row[0]*math.pi/2)*math.sin(row[1]*math.pi/2),
            (1+g)*math.sin(row[0]*math.pi/2)]
study.add_trials([optuna_amd.create_trial(params={n: float(pm[r, i]) for i, n in enumerate(names)},
                                          distributions=dists, values=dtlz2(pm[r]))
                  for r in range(n_hist)])
for _ in range(2):
    t = study.ask()
    x = np.array([t.suggest_float(n, 0, 1) for n in names])
    study.tell(t, dtlz2(x))
