from .bus import MessageBus, Topics  # noqa: F401
from .client import FederatedClient  # noqa: F401
from .coordinator import FederatedServer  # noqa: F401
from .sampler import ClientSampler  # noqa: F401
