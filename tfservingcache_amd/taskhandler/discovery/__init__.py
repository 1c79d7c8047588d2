from .base import (DiscoveryService, FileDiscovery, MockDiscovery,  # noqa: F401
                   ServingService, StaticDiscovery)
