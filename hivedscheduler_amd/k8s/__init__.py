from .client import KubeClient  # noqa: F401
from .informer import Informer  # noqa: F401
