"""Sanitizer harness: kernel determinism == LDS-race
check. Races from missing barriers manifest as run-to-run nondeterminism
under varying wave scheduling; every genrec kernel must be bitwise
deterministic (the atomics-based flash backward: fp32-tolerance instead).
Full sweep: `python tools/race_check.py --iters 20 --perturb`.
"""

import os
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "tools"))


def test_kernels_deterministic():
    from race_check import build_cases, run_case

    for name, fn, atol in build_cases():
        assert run_case(name, fn, iters=6, perturb=True, atol=atol), name
