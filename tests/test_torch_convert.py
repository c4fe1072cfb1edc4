"""Unit tests for the batch converter surface (reference
torch_dataset.py:95-236 parity): spec normalization, feature_shapes
reshaping, and the pandas object-column compatibility path."""

import numpy as np
import pandas as pd
import pytest
import torch

from ray_shuffling_data_loader_amd.torch_dataset import (
    _normalize_torch_data_spec,
    convert_to_tensor,
    dataframe_to_tensor_factory,
    rowblock_to_tensor_factory,
)
from ray_shuffling_data_loader_amd.utils.rowblock import RowBlock


def test_normalize_defaults():
    cols, shapes, types, label, lshape, ltype = _normalize_torch_data_spec(
        feature_columns="f", label_column="y"
    )
    assert cols == ["f"] and shapes == [None]
    assert types == [torch.float] and ltype == torch.float


def test_normalize_mismatch_asserts():
    with pytest.raises(AssertionError):
        _normalize_torch_data_spec(
            feature_columns=["a", "b"], feature_shapes=[1], label_column="y"
        )
    with pytest.raises(AssertionError):
        _normalize_torch_data_spec(
            feature_columns=["a"], feature_types=["not-a-dtype"],
            label_column="y",
        )


def test_feature_shapes_reshape():
    block = RowBlock(
        {
            "m": torch.arange(24, dtype=torch.float32).reshape(4, 6),
            "y": torch.arange(4, dtype=torch.float32),
        }
    )
    fn = rowblock_to_tensor_factory(
        feature_columns=["m"], feature_shapes=[(2, 3)],
        feature_types=[torch.float32], label_column="y",
    )
    feats, label = fn(block)
    assert feats[0].shape == (4, 2, 3)
    assert label.shape == (4, 1)


def test_pandas_object_column_paths():
    df = pd.DataFrame(
        {
            "arr": [np.arange(3, dtype=np.float32) for _ in range(5)],
            "lst": [[1.0, 2.0] for _ in range(5)],
            "y": np.arange(5, dtype=np.float64),
        }
    )
    feats, label = convert_to_tensor(
        df,
        feature_columns=["arr", "lst"],
        feature_shapes=[(3,), (2,)],
        feature_types=[torch.float32, torch.float32],
        label_column="y",
        label_shape=None,
        label_type=torch.float32,
    )
    assert feats[0].shape == (5, 3) and feats[1].shape == (5, 2)
    assert torch.equal(label.flatten(), torch.arange(5, dtype=torch.float32))


def test_pandas_unsupported_object_raises():
    df = pd.DataFrame({"bad": [object() for _ in range(3)],
                       "y": [0.0, 1.0, 2.0]})
    with pytest.raises(Exception, match="is not supported"):
        convert_to_tensor(
            df,
            feature_columns=["bad"],
            feature_shapes=[None],
            feature_types=[torch.float32],
            label_column="y",
            label_shape=None,
            label_type=torch.float32,
        )


def test_factory_alias():
    assert dataframe_to_tensor_factory is rowblock_to_tensor_factory
