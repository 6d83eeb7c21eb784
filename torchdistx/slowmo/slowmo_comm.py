from torchdistx_amd.slowmo.slowmo_comm import SlowMoState, slowmo_hook  # noqa: F401
