# AnyPrecisionAdamW — AdamW with user-controlled optimizer-state dtypes and
# optional Kahan-compensated weight updates, so full-bf16 training keeps
# fp32-quality weight updates.
#
# Capability parity with the reference implementation
# (/root/reference/src/python/torchdistx/optimizers/anyprecision_optimizer.py:19-182):
# same constructor surface and defaults, same state layout ("step", "exp_avg",
# "exp_avg_sq", optional "compensation"), and bitwise equivalence with
# torch.optim.AdamW when every dtype is fp32 and Kahan summation is off.
#
# MI355X-native redesign: the update is written against torch 2.10's AdamW
# op sequence (lerp_/addcmul_/addcdiv_) so the fp32 path is bitwise-identical
# to torch.optim.AdamW on this stack, and a fused single-kernel CDNA4 step
# (torchdistx_amd._K.anyprecision_adamw_) is used on ROCm GPUs when available
# instead of the ~10 separate elementwise kernels the eager path launches.

import torch
from torch.optim.optimizer import Optimizer


def _local_view(t):
    """DTensor (FSDP2) parameters carry their data in a plain contiguous
    local shard; the fused kernel runs on that shard directly — an
    elementwise optimizer step on the local shard is exactly the sharded
    step. Other tensors pass through unchanged."""
    if t is None:
        return None
    try:
        from torch.distributed.tensor import DTensor
    except ImportError:
        return t
    if isinstance(t, DTensor):
        return t.to_local()
    return t


# Fused-eligible tensors below this element count batch into one kernel
# launch per group; larger ones keep the per-tensor vectorized kernel.
_BATCH_MAX_NUMEL = 1 << 20


def _kernel_loaded() -> bool:
    try:
        from torchdistx_amd import _kernels

        return _kernels.has_anyprecision_adamw()
    except Exception:
        return False


def _fused_step_available(p: torch.Tensor) -> bool:
    p = _local_view(p)
    if p.device.type != "cuda":
        return False
    # Non-DTensor wrapper subclasses (functorch wrappers, fake modes, ...)
    # are storage-less: handing them to the raw HIP kernel would
    # dereference a null device pointer. They take the eager op sequence,
    # which dispatches through the subclass correctly.
    if torch.utils._python_dispatch.is_traceable_wrapper_subclass(p):
        return False
    return _kernel_loaded()


class AnyPrecisionAdamW(Optimizer):
    """AdamW with user-chosen state dtypes and optional Kahan summation.

    Args:
        params: iterable of parameters or dicts defining parameter groups.
        lr: learning rate (default: 1e-3).
        betas: coefficients for the running averages of the gradient and its
            square (default: (0.9, 0.999)).
        eps: denominator fuzz term (default: 1e-8).
        weight_decay: decoupled (AdamW-style) weight decay (default: 0.0).
        use_kahan_summation: keep a compensation buffer so low-precision
            weight updates accumulate with effectively higher precision
            (default: False).
        momentum_dtype: dtype of ``exp_avg`` (default: torch.float32).
        variance_dtype: dtype of ``exp_avg_sq`` (default: torch.bfloat16).
        compensation_buffer_dtype: dtype of the Kahan compensation buffer
            (default: torch.bfloat16).
        use_fused: True forces the fused single-kernel CDNA4 step and
            raises ``RuntimeError`` if it cannot run (so a benchmark can
            never silently measure the eager path); False forbids it;
            None (default) uses it automatically on GPU tensors when the
            kernel extension is loaded. DTensor (FSDP2) parameters are
            unwrapped to their local shard for the fused step. The fused
            step computes in fp32 and rounds once per state store, so it
            is close to but not bitwise-identical with the eager op
            sequence.
    """

    def __init__(
        self,
        params,
        lr=1e-3,
        betas=(0.9, 0.999),
        eps=1e-8,
        weight_decay=0.0,
        use_kahan_summation=False,
        momentum_dtype=torch.float32,
        variance_dtype=torch.bfloat16,
        compensation_buffer_dtype=torch.bfloat16,
        use_fused=None,
    ):
        defaults = dict(
            lr=lr,
            betas=betas,
            eps=eps,
            weight_decay=weight_decay,
            use_kahan_summation=use_kahan_summation,
            momentum_dtype=momentum_dtype,
            variance_dtype=variance_dtype,
            compensation_buffer_dtype=compensation_buffer_dtype,
        )
        super().__init__(params, defaults)
        self.use_fused = use_fused
        # Diagnostic: number of parameter updates that took the fused
        # CDNA4 kernel (lets tests and benchmarks assert the fast path
        # actually ran).
        self._fused_steps = 0

    def load_state_dict(self, state_dict):
        """Base-class restore, then re-cast the optimizer states to their
        configured dtypes: ``torch.optim.Optimizer.load_state_dict`` casts
        floating state tensors to the *parameter* dtype, which would
        silently destroy the any-precision state layout (e.g. a bf16
        compensation buffer next to fp32 weights)."""
        super().load_state_dict(state_dict)
        for group in self.param_groups:
            for p in group["params"]:
                state = self.state.get(p)
                if not state:
                    continue
                state["exp_avg"] = state["exp_avg"].to(group["momentum_dtype"])
                state["exp_avg_sq"] = state["exp_avg_sq"].to(
                    group["variance_dtype"]
                )
                if "compensation" in state:
                    state["compensation"] = state["compensation"].to(
                        group["compensation_buffer_dtype"]
                    )

    @torch.no_grad()
    def step(self, closure=None):
        """Performs a single optimization step."""
        if closure is not None:
            with torch.enable_grad():
                closure()

        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            lr = group["lr"]
            weight_decay = group["weight_decay"]
            eps = group["eps"]
            use_kahan = group["use_kahan_summation"]
            momentum_dtype = group["momentum_dtype"]
            variance_dtype = group["variance_dtype"]
            compensation_dtype = group["compensation_buffer_dtype"]

            # Fused-eligible small tensors accumulate here and update in
            # ONE batched kernel launch at the end of the group — a
            # transformer group is hundreds of tensors, and per-tensor
            # launch overhead would otherwise dominate the small ones.
            batch = {k: [] for k in
                     ("p", "g", "m", "v", "c", "step_size", "bc2")}

            for p in group["params"]:
                if p.grad is None:
                    continue
                if p.grad.is_sparse:
                    raise RuntimeError(
                        "AnyPrecisionAdamW does not support sparse gradients"
                    )

                state = self.state[p]
                if len(state) == 0:
                    state["step"] = torch.tensor(0.0)
                    state["exp_avg"] = torch.zeros_like(p, dtype=momentum_dtype)
                    state["exp_avg_sq"] = torch.zeros_like(p, dtype=variance_dtype)
                    if use_kahan:
                        state["compensation"] = torch.zeros_like(
                            p, dtype=compensation_dtype
                        )

                state["step"] += 1
                step = state["step"].item()
                exp_avg = state["exp_avg"]
                exp_avg_sq = state["exp_avg_sq"]
                grad = p.grad

                bias_correction1 = 1 - beta1**step
                bias_correction2_sqrt = (1 - beta2**step) ** 0.5
                step_size = lr / bias_correction1

                # Fused-path eligibility on the local shards (DTensor
                # parameters under FSDP2 unwrap to their plain contiguous
                # shard; the elementwise step on the shard IS the sharded
                # step).
                p_l = _local_view(p)
                g_l = _local_view(grad)
                m_l = _local_view(exp_avg)
                v_l = _local_view(exp_avg_sq)
                c_l = _local_view(state["compensation"]) if use_kahan else None
                can_fuse = (
                    self.use_fused is not False
                    and _fused_step_available(p)
                    and p_l.is_contiguous()
                    and g_l.is_contiguous()
                    and m_l.is_contiguous()
                    and v_l.is_contiguous()
                    and (c_l is None or c_l.is_contiguous())
                    and g_l.numel() == p_l.numel()
                    and m_l.numel() == p_l.numel()
                    and v_l.numel() == p_l.numel()
                )
                if self.use_fused is True and not can_fuse:
                    raise RuntimeError(
                        "use_fused=True, but the fused AnyPrecisionAdamW "
                        "step cannot run for this parameter (kernel "
                        "extension missing, non-GPU tensor, storage-less "
                        "wrapper subclass, or non-contiguous/mismatched "
                        "local shards). Pass use_fused=None to fall back "
                        "to the eager op sequence automatically."
                    )
                if can_fuse:
                    from torchdistx_amd import _kernels

                    self._fused_steps += 1
                    if p_l.numel() >= _BATCH_MAX_NUMEL:
                        # Big tensors take the per-tensor vectorized
                        # kernel (16-byte packed accesses).
                        _kernels.anyprecision_adamw_(
                            p_l,
                            g_l,
                            m_l,
                            v_l,
                            c_l,
                            lr,
                            beta1,
                            beta2,
                            eps,
                            weight_decay,
                            step_size,
                            bias_correction2_sqrt,
                        )
                    else:
                        batch["p"].append(p_l)
                        batch["g"].append(g_l)
                        batch["m"].append(m_l)
                        batch["v"].append(v_l)
                        batch["c"].append(c_l)
                        batch["step_size"].append(step_size)
                        batch["bc2"].append(bias_correction2_sqrt)
                    continue

                # Decoupled weight decay (AdamW).
                if weight_decay:
                    p.mul_(1 - lr * weight_decay)

                # Moment updates — the exact op sequence torch 2.10's
                # _single_tensor_adam uses, so the all-fp32 configuration is
                # bitwise-equal to torch.optim.AdamW.
                exp_avg.lerp_(grad.to(exp_avg.dtype), 1 - beta1)
                exp_avg_sq.mul_(beta2).addcmul_(
                    grad.to(exp_avg_sq.dtype), grad.to(exp_avg_sq.dtype), value=1 - beta2
                )

                denom = (exp_avg_sq.sqrt() / bias_correction2_sqrt).add_(eps)

                if use_kahan:
                    compensation = state["compensation"]
                    # Accumulate the update into the compensation buffer, then
                    # fold it into the weights, keeping the rounding error for
                    # the next step (Kahan/compensated summation).
                    compensation.addcdiv_(
                        exp_avg.to(compensation.dtype),
                        denom.to(compensation.dtype),
                        value=-step_size,
                    )
                    prev = p.detach().clone()
                    p.add_(compensation)
                    compensation.add_(prev.sub_(p))
                else:
                    p.addcdiv_(
                        exp_avg.to(p.dtype), denom.to(p.dtype), value=-step_size
                    )

            if batch["p"]:
                from torchdistx_amd import _kernels

                _kernels.anyprecision_adamw_batched_(
                    batch["p"],
                    batch["g"],
                    batch["m"],
                    batch["v"],
                    batch["c"],
                    lr,
                    beta1,
                    beta2,
                    eps,
                    weight_decay,
                    batch["step_size"],
                    batch["bc2"],
                )
