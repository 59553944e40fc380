from .provider import CloudProvider, ScaleEvent  # noqa: F401
from .fake import FakeCloudProvider  # noqa: F401
