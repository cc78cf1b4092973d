import os, sys
sys.path.insert(0, os.getcwd())
import cProfile, pstats, warnings
import numpy as np
warnings.simplefilter("ignore")
import optuna_amd
optuna_amd.logging.set_verbosity(optuna_amd.logging.WARNING)
import math

rng = np.random.RandomState(0)
n_hist = 6000
names = [f"x{i}" for i in range(10)]
dists = {n: optuna_amd.distributions.FloatDistribution(0.0, 1.0) for n in names}
sampler = optuna_amd.samplers.TPESampler(seed=0, n_startup_trials=10)
study = optuna_amd.create_study(directions=["minimize"]*3, sampler=sampler)
pm = rng.uniform(0, 1, size=(n_hist, 10))
def dtlz2(row):
    g = float(np.sum((row[2:] - 0.5) ** 2))
    return [(1+g)*math.cos(row[0]*math.pi/2)*math.cos(row[1]*math.pi/2),
            (1+g)*math.cos(row[0]*math.pi/2)*math.sin(row[1]*math.pi/2),
            (1+g)*math.sin(row[0]*math.pi/2)]
study.add_trials([
    optuna_amd.create_trial(params={n: float(pm[r, i]) for i, n in enumerate(names)},
                            distributions=dists, values=dtlz2(pm[r]))
    for r in range(n_hist)])

def one_step():
    t = study.ask()
    x = np.array([t.suggest_float(n, 0, 1) for n in names])
    study.tell(t, dtlz2(x))

one_step()
pr = cProfile.Profile(); pr.enable()
for _ in range(5):
    one_step()
pr.disable()
pstats.Stats(pr).sort_stats("cumulative").print_stats(20)
