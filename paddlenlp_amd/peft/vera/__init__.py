from .vera_model import VeRAConfig, VeRAModel  # noqa: F401
