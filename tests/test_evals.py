"""Elo / Bradley-Terry aggregation (reference `templates/evals.py:181-336`)."""

import pandas as pd

from sutro_amd.templates.evals import Rank


def test_elo_orders_transitive_rankings():
    rankings = [["a", "b", "c"]] * 20 + [["b", "a", "c"]] * 5
    out = Rank.elo(rankings)
    assert list(out.index) == ["a", "b", "c"]
    assert out.loc["a", "elo"] > out.loc["b", "elo"] > out.loc["c", "elo"]


def test_elo_mean_anchored():
    out = Rank.elo([["x", "y"]] * 10, elo_mean=1500.0)
    assert abs(out["elo"].mean() - 1500.0) < 1.0


def test_elo_from_dataframe_with_json_strings():
    df = pd.DataFrame({"ranking": ['["p","q"]', '["p","q"]', '["q","p"]']})
    out = Rank.elo(df, column="ranking")
    assert out.loc["p", "elo"] > out.loc["q", "elo"]


def test_elo_win_counts():
    out = Rank.elo([["a", "b"], ["a", "b"], ["b", "a"]])
    assert out.loc["a", "wins"] == 2
    assert out.loc["a", "losses"] == 1
    assert out.loc["a", "matches"] == 3
