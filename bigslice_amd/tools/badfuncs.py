"""Adversarial probes of the Func-registry determinism invariant
(reference cmd/badfuncs): demonstrates the failure modes the registry
digest check catches.

  python -m bigslice_amd.tools.badfuncs late     # register during run
  python -m bigslice_amd.tools.badfuncs diverge  # ranks disagree (needs
                                                 # torchrun world>1)
"""

from __future__ import annotations

import os
import sys

import torch

import bigslice_amd as bs


def late():
    """Registering a Func while a session runs must raise."""
    def build():
        # illegal: register inside a Func invocation
        bs.func(lambda: bs.Const(1, torch.arange(1)))
        return bs.Const(1, torch.arange(3, dtype=torch.int64))
    fv = bs.func(build)
    sess = bs.start(parallelism=1, device="cpu")
    try:
        sess.run(fv)
    except RuntimeError as e:
        print("caught expected registration race:", e)
        return
    print("ERROR: late registration was not detected")
    sys.exit(1)


def diverge():
    """Ranks registering different Funcs must fail the digest check."""
    rank = int(os.environ.get("RANK", "0"))
    if rank == 0:
        bs.func(lambda: bs.Const(1, torch.arange(1)))
    try:
        bs.start(distributed=True, device="cpu")
    except RuntimeError as e:
        print(f"rank {rank}: caught expected divergence: {e}")
        return
    if int(os.environ.get("WORLD_SIZE", "1")) > 1:
        print("ERROR: registry divergence was not detected")
        sys.exit(1)


def main():
    cmd = sys.argv[1] if len(sys.argv) > 1 else "late"
    if cmd == "late":
        late()
    else:
        diverge()


if __name__ == "__main__":
    main()
