import os, sys
sys.path.insert(0, os.getcwd())
import cProfile, pstats, warnings
import numpy as np
warnings.simplefilter("ignore")
import optuna_amd
optuna_amd.logging.set_verbosity(optuna_amd.logging.WARNING)

D = 20
names = [f"x{i}" for i in range(D)]
dists = {n: optuna_amd.distributions.FloatDistribution(-5.0, 5.0) for n in names}
study = optuna_amd.create_study(sampler=optuna_amd.samplers.TPESampler(seed=42, n_startup_trials=10))
rng = np.random.RandomState(0)
pm = rng.uniform(-5, 5, size=(10000, D))
vals = rng.rand(10000)
study.add_trials([
    optuna_amd.create_trial(params={n: float(pm[r, i]) for i, n in enumerate(names)},
                            distributions=dists, value=float(vals[r]))
    for r in range(10000)])

def one_step():
    t = study.ask()
    x = np.empty(D)
    for i, n in enumerate(names):
        x[i] = t.suggest_float(n, -5.0, 5.0)
    study.tell(t, float(np.sum((x - 1.0) ** 2)))

for _ in range(20):
    one_step()
pr = cProfile.Profile(); pr.enable()
for _ in range(200):
    one_step()
pr.disable()
st = pstats.Stats(pr)
st.sort_stats("tottime").print_stats(24)
