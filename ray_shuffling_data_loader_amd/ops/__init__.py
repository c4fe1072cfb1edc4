from ray_shuffling_data_loader_amd.ops.shuffle_ops import (  # noqa: F401
    gather_rows,
    pack_columns,
    partition_rows,
    unpack_permute,
)
