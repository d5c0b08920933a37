from fedtorch_amd.nodes.client import Client, Node  # noqa: F401
from fedtorch_amd.nodes.centered import (  # noqa: F401
    ClientCentered, ServerCentered)
