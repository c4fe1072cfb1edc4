from ray_shuffling_data_loader_amd.parallel import fabric  # noqa: F401
