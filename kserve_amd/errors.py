"""Data-plane error taxonomy.

Mirrors the reference's exception table wired into the REST exception handlers
(reference: python/kserve/kserve/errors.py + protocol/rest/server.py handler table).
"""


class InferenceError(RuntimeError):
    """Generic inference failure (HTTP 500)."""

    def __init__(self, reason: str, status: str = None, debug_info: str = None):
        self.reason = reason
        self.status = status
        self.debug_info = debug_info
        super().__init__(reason)

    def __str__(self):
        return self.reason


class InvalidInput(ValueError):
    """Malformed request payload (HTTP 400)."""

    def __init__(self, reason: str):
        self.reason = reason
        super().__init__(reason)


class ModelNotFound(Exception):
    """Unknown model name (HTTP 404)."""

    def __init__(self, model_name: str = None):
        self.reason = f"Model with name {model_name} does not exist."
        super().__init__(self.reason)


class ModelNotReady(Exception):
    """Model registered but not loaded/ready (HTTP 503)."""

    def __init__(self, model_name: str, detail: str = None):
        self.model_name = model_name
        self.error_msg = f"Model with name {model_name} is not ready."
        if detail:
            self.error_msg += " " + detail
        super().__init__(self.error_msg)


class ServerNotReady(Exception):
    def __init__(self, detail: str = None):
        super().__init__(detail or "Server is not ready.")


class ServerNotLive(Exception):
    def __init__(self, detail: str = None):
        super().__init__(detail or "Server is not live.")


class UnsupportedProtocol(Exception):
    def __init__(self, protocol_version: str = None):
        self.reason = f"Unsupported protocol {protocol_version}."
        super().__init__(self.reason)


class EngineDead(RuntimeError):
    """The native LLM engine loop crashed; server must fail loudly."""


class NoNativeExtension(RuntimeError):
    """Raised when a GPU tensor reaches an op whose HIP extension is missing.

    The MI355X compute path must never fall back to eager torch silently on
    a GPU box (the round-end harness records which .so files were loaded).
    """

    def __init__(self, op: str, detail: str = ""):
        super().__init__(
            f"kserve_amd native HIP extension missing for GPU op '{op}'. "
            f"Build it with `python setup.py build_ext --inplace` "
            f"(PYTORCH_ROCM_ARCH=gfx950). {detail}"
        )
