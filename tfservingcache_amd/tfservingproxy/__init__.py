from .servinghandler import LocalServingHandler, ServingError  # noqa: F401
from .rest import (make_cache_rest_app, make_proxy_rest_app,  # noqa: F401
                   parse_model_url)
from .grpc_server import (GrpcForwarder, HealthState,  # noqa: F401
                          make_cache_grpc_server, make_proxy_grpc_server)
