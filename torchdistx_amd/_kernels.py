# Loader shim for the CDNA4 kernel extension (torchdistx_amd._K).
# Importing this module registers the tdx:: init ops with the dispatcher,
# which the deferred-init replay engine redirects to on GPU targets.
# On machines without the built extension every query degrades to False and
# replay falls back to stock ATen kernels — unless TDX_REQUIRE_NATIVE_INIT
# is set, in which case the C++ redirect fails loudly.

import torch

try:
    from torchdistx_amd import _K  # noqa: F401

    _HAS_K = True
except ImportError:
    _K = None
    _HAS_K = False


def available() -> bool:
    return _HAS_K


def has_init_kernels() -> bool:
    return _HAS_K


def has_anyprecision_adamw() -> bool:
    return _HAS_K and torch.cuda.is_available()


def anyprecision_adamw_(
    param,
    grad,
    exp_avg,
    exp_avg_sq,
    compensation,
    lr,
    beta1,
    beta2,
    eps,
    weight_decay,
    step_size,
    bias_correction2_sqrt,
):
    _K.anyprecision_adamw_(
        param,
        grad.contiguous(),
        exp_avg,
        exp_avg_sq,
        compensation,
        lr,
        beta1,
        beta2,
        eps,
        weight_decay,
        step_size,
        bias_correction2_sqrt,
    )


def anyprecision_adamw_batched_(
    params,
    grads,
    exp_avgs,
    exp_avg_sqs,
    compensations,
    lr,
    beta1,
    beta2,
    eps,
    weight_decay,
    step_sizes,
    bias_correction2_sqrts,
):
    """One kernel launch updating every listed parameter (see
    csrc/hip/anyprecision_adamw.hip, adamw_batched_kernel)."""
    _K.anyprecision_adamw_batched_(
        params,
        [g.contiguous() for g in grads],
        exp_avgs,
        exp_avg_sqs,
        compensations,
        lr,
        beta1,
        beta2,
        eps,
        weight_decay,
        step_sizes,
        bias_correction2_sqrts,
    )
