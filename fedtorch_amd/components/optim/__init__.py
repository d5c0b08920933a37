from fedtorch_amd.components.optim.sgd import FusedSGD  # noqa: F401
from fedtorch_amd.components.optim.adam import FusedAdamW  # noqa: F401
