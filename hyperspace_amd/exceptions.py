"""Hyperspace-AMD exceptions.

Mirrors the reference exception surface
(/root/reference/src/main/scala/com/microsoft/hyperspace/HyperspaceException.scala).
"""


class HyperspaceException(Exception):
    """Generic user-facing error raised by the engine."""


class NoChangesException(HyperspaceException):
    """Internal no-op signal: a refresh/optimize found nothing to do.

    Reference: actions/NoChangesException.scala — caught in Action.run and
    turned into a clean abort of the transaction.
    """


class KernelUnavailableError(HyperspaceException):
    """Raised when a HIP kernel path is required (device tensors on a GPU box)
    but the native extension is not loaded.  GPU execution never silently
    falls back to eager PyTorch."""
