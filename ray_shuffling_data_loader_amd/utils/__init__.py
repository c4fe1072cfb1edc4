from ray_shuffling_data_loader_amd.utils.rowblock import RowBlock  # noqa: F401
from ray_shuffling_data_loader_amd.utils.schema import ColumnSpec, Schema  # noqa: F401
