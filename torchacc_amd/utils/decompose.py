"""Decomposition-table shim (reference utils/decompose.py:23-128).

The reference rewrote torch's global decomposition table so in-place ops
traced cleanly through LazyTensor. The eager MI355X backend executes aten
ops directly — there is no trace to keep clean — so this is a documented
no-op preserved for API compatibility.
"""


def replace_decompose() -> None:
    """No-op on the eager backend."""
