from fedtorch_amd.nodes.client import Client  # noqa: F401
