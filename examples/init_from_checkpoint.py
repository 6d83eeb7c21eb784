"""Pretrained-weights flow: build the model FAKE (no allocation, no init
compute), then load a checkpoint with assign=True — the recorded init
work is simply never executed. Anything the checkpoint does not cover
materializes from the tape afterwards, bitwise-equal (on the
pinned-Philox native path) to what a full materialization would have
produced.

Run on one MI355X:
    python examples/init_from_checkpoint.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from torchdistx_amd import deferred_init, is_deferred
from torchdistx_amd.deferred_init import materialize_module
from torchdistx_amd.models import LLAMA3_8B, TINY, build_model


def main() -> None:
    device = "cuda" if torch.cuda.is_available() else "cpu"
    cfg = LLAMA3_8B if device == "cuda" else TINY
    dtype = torch.bfloat16 if device == "cuda" else torch.float32

    # Pretend this checkpoint exists: here it is synthesized from a
    # second (materialized) copy; in production it comes from
    # torch.load / safetensors with the same keys.
    torch.manual_seed(7)
    donor = deferred_init(build_model, cfg, device=device, dtype=dtype)
    materialize_module(donor)
    ckpt = {k: v for k, v in donor.state_dict().items()}
    del donor

    # The actual flow: fake build (instant, no HBM), load, done.
    torch.manual_seed(123)  # irrelevant: init never runs for loaded keys
    model = deferred_init(build_model, cfg, device=device, dtype=dtype)
    assert is_deferred(model)
    model.load_state_dict(ckpt, assign=True)
    # Non-persistent buffers (e.g. RoPE caches) are not in the state
    # dict; they materialize from the tape.
    materialize_module(model)
    assert not is_deferred(model)

    tokens = torch.randint(0, cfg.vocab_size, (2, 32), device=device)
    loss = model.loss(tokens)
    print(f"loaded {sum(p.numel() for p in model.parameters()):,} params; "
          f"loss {loss.item():.3f}")


if __name__ == "__main__":
    main()
