from torchdistx_amd.slowmo.slowmo_optimizer import SlowMomentumOptimizer  # noqa: F401
