from ray_shuffling_data_loader_amd.ops.shuffle_ops import (  # noqa: F401
    gather_rows,
    pack_columns,
    partition_rows,
    relu_mask_words,
    t_frag_swizzle,
    t_frag_unswizzle,
    unpack_permute,
    wgrad,
    wgrad_frag,
)
