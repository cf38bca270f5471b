from sentio_amd.models.document import Document  # noqa: F401
