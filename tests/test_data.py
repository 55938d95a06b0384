import numpy as np
import pytest

from spark_rapids_ml_amd.data import DataFrame, extract_features


def test_from_numpy_and_columns():
    X = np.random.rand(10, 4).astype(np.float32)
    y = np.arange(10.0)
    df = DataFrame.from_numpy(X, y)
    assert df.columns == ["features", "label"]
    assert df.num_rows == 10
    assert np.allclose(df["features"], X)


def test_extract_features_vector_col():
    X = np.random.rand(8, 3).astype(np.float64)
    df = DataFrame.from_numpy(X)
    out = extract_features(df, "features", None, float32_inputs=True)
    assert out.dtype == np.float32
    out64 = extract_features(df, "features", None, float32_inputs=False)
    assert out64.dtype == np.float64


def test_extract_features_multi_cols():
    df = DataFrame({"a": np.arange(5.0), "b": np.arange(5.0) * 2})
    out = extract_features(df, None, ["a", "b"], True)
    assert out.shape == (5, 2)
    assert np.allclose(out[:, 1], np.arange(5) * 2)


def test_with_column_select_drop():
    df = DataFrame({"a": np.arange(4)})
    df2 = df.with_column("b", np.ones(4))
    assert set(df2.columns) == {"a", "b"}
    assert df2.select("b").columns == ["b"]
    assert df2.drop("a").columns == ["b"]


def test_row_mismatch_raises():
    with pytest.raises(ValueError):
        DataFrame({"a": np.arange(3), "b": np.arange(4)})


def test_parquet_roundtrip(tmp_path):
    X = np.random.rand(20, 4).astype(np.float32)
    y = np.random.rand(20)
    df = DataFrame.from_numpy(X, y)
    path = str(tmp_path / "data")
    df.write_parquet(path)
    back = DataFrame.read_parquet(path, vector_cols=["features"])
    assert back.num_rows == 20
    assert np.allclose(np.asarray(back["features"]), X)
    assert np.allclose(np.asarray(back["label"]), y)


def test_to_pandas():
    X = np.random.rand(6, 2).astype(np.float32)
    df = DataFrame.from_numpy(X)
    pdf = df.to_pandas()
    assert len(pdf) == 6
    assert np.allclose(np.stack(pdf["features"].to_list()), X)
