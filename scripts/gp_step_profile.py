import os, sys
sys.path.insert(0, os.getcwd())
import cProfile, pstats, warnings
import numpy as np
warnings.simplefilter("ignore")
import optuna_amd
optuna_amd.logging.set_verbosity(optuna_amd.logging.WARNING)

n_obs, D = 5000, 20
names = [f"x{i}" for i in range(D)]
dists = {n: optuna_amd.distributions.FloatDistribution(-5.0, 5.0) for n in names}
study = optuna_amd.create_study(sampler=optuna_amd.samplers.GPSampler(seed=0, n_startup_trials=10))
rng = np.random.RandomState(0)
pm = rng.uniform(-5, 5, size=(n_obs, D))
vals = rng.rand(n_obs)
study.add_trials([optuna_amd.create_trial(params={n: float(pm[r, i]) for i, n in enumerate(names)},
                                          distributions=dists, value=float(vals[r]))
                  for r in range(n_obs)])

def one_step():
    t = study.ask()
    x = np.array([t.suggest_float(n, -5, 5) for n in names])
    study.tell(t, float(np.sum((x - 1) ** 2)))

for _ in range(3):
    one_step()
pr = cProfile.Profile(); pr.enable()
for _ in range(12):
    one_step()
pr.disable()
pstats.Stats(pr).sort_stats("cumulative").print_stats(26)
