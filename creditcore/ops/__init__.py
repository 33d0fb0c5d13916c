"""Scoring ops: HIP-kernel wrappers + CPU reference implementations.

``creditcore.ops.gpu`` loads the in-tree HIP extension (csrc/). On a GPU box
the extension is REQUIRED — scoring ops raise if it is missing rather than
silently falling back to an eager path. ``creditcore.ops.cpu_ref`` is the
fp32 CPU reference every kernel is tested against.
"""
