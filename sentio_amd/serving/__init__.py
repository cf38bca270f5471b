from sentio_amd.serving.app import create_app  # noqa: F401
from sentio_amd.serving.container import ServiceContainer, get_container  # noqa: F401
