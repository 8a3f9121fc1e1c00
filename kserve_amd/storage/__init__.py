from kserve_amd.storage.storage import Storage  # noqa: F401
