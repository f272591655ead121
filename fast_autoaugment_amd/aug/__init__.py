"""Augmentation subsystem.

The reference applies 19 PIL ops on CPU inside DataLoader workers
(reference augmentations.py:13-182, data.py:253-264). Here the whole
pipeline is GPU-resident: for each batch the host compiles a per-image
"op program" (all RNG drawn up front — sub-policy pick, Bernoulli gates,
sign mirrors, crop/flip/cutout coordinates), and a single HIP kernel
executes the program per image (LDS-resident for 32x32, global ping-pong
for ImageNet sizes), finishing with pad-crop, horizontal flip, normalize to
bf16 NHWC and post-normalize cutout.

The CPU executor (`cpu_exec`) interprets the same programs with numpy and
is the numerics reference for the HIP kernel; it is itself golden-tested
against PIL in tests/test_aug_pil_golden.py.
"""
from .ops import (OpCode, OP_RANGES, compile_program, compile_post, PROG_SLOTS, PROG_WIDTH)
from .cpu_exec import apply_program_batch, apply_post_batch, run_pipeline_cpu

__all__ = [
    "OpCode", "OP_RANGES", "compile_program", "compile_post",
    "PROG_SLOTS", "PROG_WIDTH",
    "apply_program_batch", "apply_post_batch", "run_pipeline_cpu",
]
