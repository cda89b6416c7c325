from .servers import grad_server_helper, model_server_helper

__all__ = ["model_server_helper", "grad_server_helper"]
