from fedtorch_amd.trainings.centered.main import (  # noqa: F401
    train_and_validate_federated_centered)
