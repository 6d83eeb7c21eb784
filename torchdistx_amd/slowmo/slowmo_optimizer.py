# Slow Momentum optimizer wrapper (arXiv:1910.00643).
#
# Capability parity with the reference
# (/root/reference/src/python/torchdistx/slowmo/slowmo_optimizer.py:11-235):
# wraps an arbitrary torch.optim.Optimizer; every step runs the base
# optimizer and a PeriodicModelAverager (global exact parameter averaging
# every ``slowmo_freq`` steps); after each averaging round applies the slow
# ("outer") momentum update
#     m   <- slowmo_factor * m + (prev - param) / lr
#     prev <- prev - slowmo_lr * lr * m
#     param <- prev
# State-dict layout matches the reference: the base optimizer dict plus
# "slowmo_freq" / "slowmo_factor" / "slowmo_lr" / "step" entries
# (reference slowmo_optimizer.py:156-189). ``_prev_parameters`` is
# deliberately NOT checkpointed (reconstructed at construction), matching
# reference slowmo_optimizer.py:132-144.
#
# Differences from the reference (intentional fixes):
#   * the slow-momentum buffer is allocated on the parameter's device rather
#     than unconditionally on the current CUDA device, so the optimizer also
#     works on CPU (gloo) test rigs and mixed-device param groups.

import torch
import torch.distributed.algorithms.model_averaging.averagers as averagers


def _check_groups_have_lr(optim: torch.optim.Optimizer) -> None:
    """The slow-momentum update divides by each group's lr, so every group
    must carry one (and the optimizer must have groups at all)."""
    if not optim.param_groups:
        raise ValueError("the base optimizer has no parameter groups")
    if any("lr" not in g for g in optim.param_groups):
        raise ValueError(
            "every parameter group needs an explicit learning rate: the "
            "slow-momentum outer update rescales by 1/lr per group"
        )


class SlowMomentumOptimizer(torch.optim.Optimizer):
    """Wraps a base optimizer and runs distributed training with Slow
    Momentum. Designed for FSDP modules with a ``NO_SHARD`` strategy together
    with :func:`torchdistx_amd.slowmo.slowmo_hook`.

    Args:
        base_optim: the base optimizer updating the local model replica.
        slowmo_freq: run parameter averaging + slow momentum every this many
            steps (default: 48).
        slowmo_factor: slow momentum coefficient (default: 0.5).
        slowmo_lr: slow momentum learning-rate scale (default: 1.0).
    """

    def __init__(
        self,
        base_optim: torch.optim.Optimizer,
        slowmo_freq: int = 48,
        slowmo_factor: float = 0.5,
        slowmo_lr: float = 1.0,
    ):
        if base_optim is None:
            raise ValueError("a base optimizer is required (got None)")
        self._base_optim = base_optim
        _check_groups_have_lr(base_optim)
        self.param_groups = self._base_optim.param_groups

        if slowmo_freq < 1:
            raise ValueError(f"slowmo_freq must be >= 1, got {slowmo_freq}")
        self.slowmo_freq = slowmo_freq
        if slowmo_factor < 0.0:
            raise ValueError(
                f"slowmo_factor must be >= 0, got {slowmo_factor}"
            )
        self.slowmo_factor = slowmo_factor
        if slowmo_lr < 0.0:
            raise ValueError(f"slowmo_lr must be >= 0, got {slowmo_lr}")
        self.slowmo_lr = slowmo_lr

        self.averager = averagers.PeriodicModelAverager(
            period=slowmo_freq, warmup_steps=0
        )

        # Snapshot of the parameters at the last slow-momentum update, in
        # flattened param_groups order. Kept outside ``self.state`` because
        # many base optimizers use an empty per-param state as the "first
        # step" signal.
        self._prev_parameters = [
            param.detach().clone()
            for group in self.param_groups
            for param in group["params"]
        ]

    @property
    def state(self):
        """Forwards to the base optimizer's ``state``."""
        return self._base_optim.state

    def __repr__(self):
        return self._base_optim.__repr__()

    def state_dict(self):
        """Base optimizer ``state_dict`` plus the SlowMo hyper-parameters and
        the averager step counter."""
        sd = self._base_optim.state_dict()
        sd["slowmo_freq"] = self.slowmo_freq
        sd["slowmo_factor"] = self.slowmo_factor
        sd["slowmo_lr"] = self.slowmo_lr
        sd["step"] = self.averager.step
        return sd

    def load_state_dict(self, state_dict):
        """Restores the base optimizer state and the SlowMo entries written
        by :meth:`state_dict`."""
        state_dict = dict(state_dict)
        self.slowmo_freq = state_dict.pop("slowmo_freq")
        self.averager.period = self.slowmo_freq
        self.slowmo_factor = state_dict.pop("slowmo_factor")
        self.slowmo_lr = state_dict.pop("slowmo_lr")
        self.averager.step = state_dict.pop("step")
        self._base_optim.load_state_dict(state_dict)
        _check_groups_have_lr(self._base_optim)

    @torch.no_grad()
    def step(self):
        """One local step; every ``slowmo_freq`` steps a global parameter
        average followed by the slow momentum update."""
        self._base_optim.step()
        # The averager all-reduce-averages the parameters every
        # ``slowmo_freq`` calls and increments its step counter every call.
        self.averager.average_parameters(params=self.param_groups)
        just_averaged = (
            self.averager.step != 1
            and (self.averager.step - 1) % self.slowmo_freq == 0
        )
        if not just_averaged:
            return

        idx = 0
        for group in self.param_groups:
            inv_lr = 1.0 / group["lr"]
            for param in group["params"]:
                pstate = self.state[param]
                if "slow_momentum" not in pstate:
                    pstate["slow_momentum"] = torch.zeros(
                        param.shape, device=param.device
                    )
                momentum = pstate["slow_momentum"]
                prev = self._prev_parameters[idx]
                # m <- factor*m + (prev - param)/lr
                momentum.mul_(self.slowmo_factor).sub_(param, alpha=inv_lr).add_(
                    prev, alpha=inv_lr
                )
                # prev <- prev - slowmo_lr*lr*m ; param <- prev
                prev.add_(momentum, alpha=-self.slowmo_lr * group["lr"])
                param.copy_(prev)
                idx += 1

    def zero_grad(self, set_to_none: bool = False):  # type: ignore[override]
        self._base_optim.zero_grad(set_to_none=set_to_none)

    def add_param_group(self, param_group):
        self._base_optim.add_param_group(param_group)
        for param in param_group["params"]:
            self._prev_parameters.append(param.detach().clone())
