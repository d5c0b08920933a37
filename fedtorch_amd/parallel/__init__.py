from fedtorch_amd.parallel.arena import Arena  # noqa: F401
