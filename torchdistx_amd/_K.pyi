# Typing stub for the CDNA4 kernel extension.

from typing import List, Optional

import torch

def has_init_kernels() -> bool: ...
def has_anyprecision_adamw() -> bool: ...
def has_anyprecision_adamw_batched() -> bool: ...
def has_batched_init() -> bool: ...
def anyprecision_adamw_(
    param: torch.Tensor,
    grad: torch.Tensor,
    exp_avg: torch.Tensor,
    exp_avg_sq: torch.Tensor,
    compensation: Optional[torch.Tensor],
    lr: float,
    beta1: float,
    beta2: float,
    eps: float,
    weight_decay: float,
    step_size: float,
    bias_correction2_sqrt: float,
) -> None: ...
def anyprecision_adamw_batched_(
    params: List[torch.Tensor],
    grads: List[torch.Tensor],
    exp_avgs: List[torch.Tensor],
    exp_avg_sqs: List[torch.Tensor],
    compensations: List[Optional[torch.Tensor]],
    lr: float,
    beta1: float,
    beta2: float,
    eps: float,
    weight_decay: float,
    step_sizes: List[float],
    bias_correction2_sqrts: List[float],
) -> None: ...
def batched_init_(
    tensors: List[torch.Tensor],
    dists: List[int],
    p0s: List[float],
    p1s: List[float],
    seeds: List[int],
    offsets: List[int],
) -> None: ...
