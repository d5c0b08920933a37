from fedtorch_amd.trainings.local_sgd import train_and_validate  # noqa: F401
from fedtorch_amd.trainings.federated import (  # noqa: F401
    train_and_validate_federated)
