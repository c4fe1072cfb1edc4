"""TorchShufflingDataset: PyTorch IterableDataset over the shuffling loader.

API parity with the reference (reference: ray_shuffling_data_loader/
torch_dataset.py:14-236): same constructor (feature_columns/shapes/types,
label_column/shape/type with identical defaults and normalization), yields
``(List[Tensor], Tensor)`` per batch. Differences by design:

  * Batches arrive as GPU-resident RowBlocks from the MI355X shuffle engine;
    the per-column cast/pack that the reference does on CPU with
    ``torch.as_tensor`` per batch (torch_dataset.py:204-236) happened ONCE on
    the GPU inside the fused reducer kernel — here we only take views /
    cheap casts.
  * ``feature_matrix=True`` (MI355X extension): when the engine produced a
    fused feature matrix column, yields ``([features_matrix], labels)`` with
    zero per-batch work — the flagship benchmark path.
"""

import functools
from typing import Any, Callable, Iterable, List, Optional, Tuple

import numpy as np
import torch
from torch.utils.data import IterableDataset

from ray_shuffling_data_loader_amd.dataset import ShufflingDataset
from ray_shuffling_data_loader_amd.utils.rowblock import RowBlock


class TorchShufflingDataset(IterableDataset):
    """A PyTorch shuffling dataset that yields (features, label) tensor
    batches upon iteration; thin wrapper around ShufflingDataset
    (reference torch_dataset.py:14-92)."""

    def __init__(
        self,
        filenames: List[str],
        num_epochs: int,
        num_trainers: int,
        batch_size: int,
        rank: int,
        drop_last: bool = False,
        num_reducers: Optional[int] = None,
        max_concurrent_epochs: int = 2,
        feature_columns: List[Any] = None,
        feature_shapes: Optional[List[Any]] = None,
        feature_types: Optional[List[torch.dtype]] = None,
        label_column: Any = None,
        label_shape: Optional[int] = None,
        label_type: Optional[torch.dtype] = None,
        feature_matrix: bool = False,
        **engine_kwargs,
    ):
        super().__init__()
        self._feature_matrix = feature_matrix
        if feature_matrix:
            # Fuse the feature columns into one [N, C] matrix inside the
            # engine's packed layout (io.fuse_schema): zero per-batch conversion.
            engine_kwargs.setdefault(
                "feature_matrix", ("__features__", list(feature_columns))
            )
            # A requested non-fp32 feature dtype (e.g. bf16 for MFMA compute)
            # is produced by the fused unpack+cast kernel, not per batch.
            if feature_types and isinstance(feature_types, list):
                ft = feature_types[0]
                if isinstance(ft, torch.dtype) and ft != torch.float32:
                    engine_kwargs.setdefault(
                        "out_dtypes", {"__features__": ft}
                    )
        self._ds = ShufflingDataset(
            filenames,
            num_epochs,
            num_trainers,
            batch_size,
            rank,
            drop_last=drop_last,
            num_reducers=num_reducers,
            max_concurrent_epochs=max_concurrent_epochs,
            **engine_kwargs,
        )
        self._batch_transform = rowblock_to_tensor_factory(
            feature_columns=feature_columns,
            feature_shapes=feature_shapes,
            feature_types=feature_types,
            label_column=label_column,
            label_shape=label_shape,
            label_type=label_type,
            feature_matrix=feature_matrix,
        )

    def set_epoch(self, epoch):
        """Set the current training epoch; call before constructing the
        iterator each epoch (reference torch_dataset.py:78-88)."""
        self._ds.set_epoch(epoch)

    def __iter__(self):
        for block in iter(self._ds):
            yield self._batch_transform(block)


def _normalize_torch_data_spec(
    feature_columns: List[Any] = None,
    feature_shapes: Optional[List[Any]] = None,
    feature_types: Optional[List[torch.dtype]] = None,
    label_column: Any = None,
    label_shape: Optional[int] = None,
    label_type: Optional[torch.dtype] = None,
):
    """Defaults + validation, identical rules to the reference
    (torch_dataset.py:144-201): feature_types default torch.float, shapes
    default None, label_type default torch.float."""
    if not isinstance(feature_columns, list):
        feature_columns = [feature_columns]

    if feature_shapes:
        if not isinstance(feature_shapes, list):
            feature_shapes = [feature_shapes]
        assert len(feature_columns) == len(
            feature_shapes
        ), "The feature_shapes size must match the feature_columns"
        for i in range(len(feature_shapes)):
            if not isinstance(feature_shapes[i], Iterable):
                feature_shapes[i] = [feature_shapes[i]]
    else:
        feature_shapes = [None] * len(feature_columns)

    if feature_types:
        if not isinstance(feature_types, list):
            feature_types = [feature_types]
        assert len(feature_columns) == len(
            feature_types
        ), "The feature_types size must match the feature_columns"
        assert all(
            isinstance(dtype, torch.dtype) for dtype in feature_types
        ), "All value in feature_types should be torch.dtype instance"
    else:
        feature_types = [torch.float] * len(feature_columns)

    if not label_type:
        label_type = torch.float

    return (
        feature_columns,
        feature_shapes,
        feature_types,
        label_column,
        label_shape,
        label_type,
    )


def rowblock_to_tensor_factory(
    feature_columns: List[Any] = None,
    feature_shapes: Optional[List[Any]] = None,
    feature_types: Optional[List[torch.dtype]] = None,
    label_column: Any = None,
    label_shape: Optional[int] = None,
    label_type: Optional[torch.dtype] = None,
    feature_matrix: bool = False,
) -> Callable[[RowBlock], Tuple[List[torch.Tensor], torch.Tensor]]:
    """RowBlock -> (List[Tensor], Tensor) converter (the reference's
    dataframe_to_tensor_factory, torch_dataset.py:95-141)."""
    (
        feature_columns,
        feature_shapes,
        feature_types,
        label_column,
        label_shape,
        label_type,
    ) = _normalize_torch_data_spec(
        feature_columns,
        feature_shapes,
        feature_types,
        label_column,
        label_shape,
        label_type,
    )
    return functools.partial(
        convert_to_tensor,
        feature_columns=feature_columns,
        feature_shapes=feature_shapes,
        feature_types=feature_types,
        label_column=label_column,
        label_shape=label_shape,
        label_type=label_type,
        feature_matrix=feature_matrix,
    )


# Back-compat alias matching the reference factory name.
dataframe_to_tensor_factory = rowblock_to_tensor_factory


def _col_tensor(block, col) -> torch.Tensor:
    if isinstance(block, RowBlock):
        return block[col]
    # pandas DataFrame compatibility (reference torch_dataset.py:211-221)
    column = block[col].values
    if column.dtype == np.object_:
        if isinstance(column[0], np.ndarray):
            column = np.stack(column)
        elif isinstance(column[0], (list, tuple)):
            column = np.array(list(column))
        else:
            raise Exception(
                f"Column {col}'s type: {type(column[0])} is not supported."
                " It must be numpy built in type or numpy object of "
                "(ndarray, list, tuple)"
            )
    return torch.as_tensor(np.ascontiguousarray(column))


def convert_to_tensor(
    block,
    feature_columns: List[Any],
    feature_shapes: List[Any],
    feature_types: List[torch.dtype],
    label_column: Any,
    label_shape: Optional[int],
    label_type: torch.dtype,
    feature_matrix: bool = False,
) -> Tuple[List[torch.Tensor], torch.Tensor]:
    """Per-batch conversion (reference torch_dataset.py:204-236). On the
    MI355X path the heavy cast/pack already happened in the fused reducer
    kernel, so this is views + dtype adjustments only."""
    if feature_matrix and isinstance(block, RowBlock) and (
        "__features__" in block.columns
    ):
        feats = block["__features__"]
        if feature_types and feats.dtype != feature_types[0]:
            feats = feats.to(feature_types[0])
        label = block[label_column].to(label_type)
        label = label.reshape(-1, label_shape if label_shape else 1)
        return [feats], label

    feature_tensor = []
    for col, shape, dtype in zip(
        feature_columns, feature_shapes, feature_types
    ):
        t = _col_tensor(block, col)
        t = t.to(dtype)
        if shape is not None:
            t = t.reshape(-1, *shape)
        else:
            t = t.reshape(-1, 1)
        feature_tensor.append(t)

    label_tensor = _col_tensor(block, label_column).to(label_type)
    if label_shape:
        label_tensor = label_tensor.reshape(-1, label_shape)
    else:
        label_tensor = label_tensor.reshape(-1, 1)
    return feature_tensor, label_tensor


if __name__ == "__main__":
    import shutil
    import tempfile

    from ray_shuffling_data_loader_amd.data_generation import (
        DATA_SPEC,
        generate_data,
    )
    from ray_shuffling_data_loader_amd.utils.schema import (
        NUMPY_TO_TORCH_DTYPE,
    )
    from ray_shuffling_data_loader_amd.utils.stats import human_readable_size

    num_rows = 10**6
    num_files = 10
    data_dir = tempfile.mkdtemp()
    print(f"Generating {num_rows} rows over {num_files} files.")
    filenames, num_bytes = generate_data(
        num_rows, num_files, 1, 0.0, data_dir
    )
    print(
        f"Generated {len(filenames)} files containing {num_rows} rows "
        f"totalling {human_readable_size(num_bytes)}."
    )
    num_epochs = 4
    batch_size = 20000
    num_reducers = 8
    feature_columns = list(DATA_SPEC.keys())
    feature_types = [
        NUMPY_TO_TORCH_DTYPE[np.dtype(dt)] for _, _, dt in DATA_SPEC.values()
    ]
    label_column = feature_columns.pop()
    label_type = feature_types.pop()
    ds = TorchShufflingDataset(
        list(filenames),
        num_epochs,
        1,
        batch_size,
        0,
        num_reducers=num_reducers,
        max_concurrent_epochs=2,
        feature_columns=feature_columns,
        feature_types=feature_types,
        label_column=label_column,
        label_type=label_type,
    )
    for epoch in range(num_epochs):
        ds.set_epoch(epoch)
        for batch_idx, (data, targets) in enumerate(ds):
            print(
                f"Epoch {epoch} - consuming batch {batch_idx}: "
                f"{len(data)} features, {len(targets)} samples"
            )
    print("Done consuming batches.")
    shutil.rmtree(data_dir)
