"""Compute ops: golden eager functional forms + the gfx950 HIP extension.

``functional`` is the eager/golden path (CPU and reference numerics).
``hip`` wraps the compiled extension (r2d2_hip) when present.  On a GPU box
the HIP path is mandatory for the ops it implements — wrappers raise if the
extension is missing rather than silently falling back to eager.
"""

from . import functional  # noqa: F401
