"""cProfile one MO-TPE suggest loop at 6k history (GPU box)."""
import os, sys
sys.path.insert(0, os.getcwd())
import cProfile, pstats, warnings
import numpy as np
warnings.simplefilter("ignore")
import optuna_amd
optuna_amd.logging.set_verbosity(optuna_amd.logging.WARNING)

rng = np.random.RandomState(0)
n_hist = 6000
names = [f"x{i}" for i in range(10)]
dists = {n: optuna_amd.distributions.FloatDistribution(0.0, 1.0) for n in names}
sampler = optuna_amd.samplers.TPESampler(seed=0, n_startup_trials=10)
study = optuna_amd.create_study(directions=["minimize", "minimize"], sampler=sampler)
pm = rng.uniform(0, 1, size=(n_hist, 10))
study.add_trials([
    optuna_amd.create_trial(
        params={n: float(pm[r, i]) for i, n in enumerate(names)},
        distributions=dists,
        values=[float(pm[r, 0]), float(1.0 - pm[r, 0] + 0.1 * pm[r, 1])],
    ) for r in range(n_hist)
])

def one_step():
    t = study.ask()
    x = np.array([t.suggest_float(n, 0, 1) for n in names])
    f1 = float(x[0]); g = 1.0 + 9.0 * float(np.mean(x[1:]))
    study.tell(t, (f1, g * (1.0 - (f1 / g) ** 0.5)))

for _ in range(3):
    one_step()
pr = cProfile.Profile()
pr.enable()
for _ in range(30):
    one_step()
pr.disable()
st = pstats.Stats(pr)
st.sort_stats("cumulative").print_stats(28)
