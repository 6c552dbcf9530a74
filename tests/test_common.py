"""Input preparation + schema normalization (mirrors reference
`tests/test_sdk.py` TestPrepareInputData patterns)."""

import pandas as pd
import pyarrow as pa
import pytest
from pydantic import BaseModel

from sutro_amd.common import (
    do_dataframe_column_concatenation,
    normalize_output_schema,
    prepare_input_data,
    to_colored_text,
)


def test_list_passthrough():
    assert prepare_input_data(["a", "b"]) == ["a", "b"]


def test_dataframe_column():
    df = pd.DataFrame({"x": ["p", "q"], "y": [1, 2]})
    assert prepare_input_data(df, "x") == ["p", "q"]


def test_dataframe_requires_column():
    with pytest.raises(ValueError):
        prepare_input_data(pd.DataFrame({"x": [1]}))


def test_dataframe_missing_column():
    with pytest.raises(ValueError):
        prepare_input_data(pd.DataFrame({"x": [1]}), "nope")


def test_multi_column_concat_with_separators():
    df = pd.DataFrame({"a": ["1", "2"], "b": ["x", "y"]})
    out = do_dataframe_column_concatenation(df, ["a", " - ", "b"])
    assert out == ["1 - x", "2 - y"]


def test_concat_via_prepare_input():
    df = pd.DataFrame({"a": ["1"], "b": ["x"]})
    assert prepare_input_data(df, ["a", "|", "b"]) == ["1|x"]


def test_arrow_table():
    t = pa.table({"c": ["u", "v"]})
    assert prepare_input_data(t, "c") == ["u", "v"]


def test_dataset_id_passthrough():
    assert prepare_input_data("dataset-abc123", "col") == "dataset-abc123"


def test_url_passthrough():
    url = "https://example.com/data.csv"
    assert prepare_input_data(url, "col") == url


def test_csv_file(tmp_path):
    p = tmp_path / "d.csv"
    pd.DataFrame({"t": ["hello", "world"]}).to_csv(p, index=False)
    assert prepare_input_data(str(p), "t") == ["hello", "world"]


def test_parquet_file(tmp_path):
    p = tmp_path / "d.parquet"
    pd.DataFrame({"t": ["a"]}).to_parquet(p)
    assert prepare_input_data(str(p), "t") == ["a"]


def test_txt_file(tmp_path):
    p = tmp_path / "d.txt"
    p.write_text("l1\nl2\n\n")
    assert prepare_input_data(str(p)) == ["l1", "l2"]


def test_unsupported_file(tmp_path):
    with pytest.raises(ValueError):
        prepare_input_data(str(tmp_path / "d.xlsx"), "c")


def test_schema_dict_passthrough():
    s = {"type": "object", "properties": {}}
    assert normalize_output_schema(s) is s


def test_schema_pydantic():
    class M(BaseModel):
        a: int

    s = normalize_output_schema(M)
    assert s["properties"]["a"]["type"] == "integer"


def test_schema_invalid():
    with pytest.raises(TypeError):
        normalize_output_schema("not a schema")


def test_colored_text_states():
    # non-tty (test env): plain passthrough
    assert "hello" in to_colored_text("hello", "success")


def test_tokenizer_roundtrip_property():
    """Property: the BPE tokenizer round-trips ANY unicode text exactly at
    every truncation (byte fallback guarantees coverage)."""
    from hypothesis import given, settings, strategies as st

    from sutro_amd.engine.tokenizer import get_tokenizer

    toks = [get_tokenizer(), get_tokenizer(512), get_tokenizer(32000)]

    @settings(max_examples=200, deadline=None)
    @given(st.text(max_size=200))
    def run(text):
        for tok in toks:
            ids = tok.encode(text)
            assert tok.decode(ids) == text
            assert all(0 <= t < tok.vocab_size for t in ids)

    run()


def test_tokenizer_token_bytes_consistent():
    """decode == concat(token_bytes) and multi-byte merges really exist."""
    from sutro_amd.engine.tokenizer import get_tokenizer

    tok = get_tokenizer()
    s = 'The quick {"name": "value", "score": 42} fox.'
    ids = tok.encode(s)
    assert b"".join(tok.token_bytes(i) for i in ids).decode() == s
    assert any(len(tok.token_bytes(i)) > 1 for i in ids)  # real BPE merges
    assert len(ids) < len(s.encode())  # compresses vs bytes
