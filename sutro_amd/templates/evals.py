"""score()/rank()/elo(): LLM-as-judge evaluation templates.

Behavioral parity with reference `templates/evals.py`:
- score(): integer scoring in a range via JSON-schema guided output.
- rank(): listwise ranking of labeled options; optional Elo aggregation.
- Rank.elo(): client-side Bradley-Terry ratings via MM iteration
  (Hunter 2004), Laplace smoothing, ties counted 0.5, scaled to an Elo
  distribution with a chosen mean (reference `evals.py:181-336`).
"""

from __future__ import annotations

import json
from itertools import combinations
from typing import List, Optional, Tuple, Union

import numpy as np
import pandas as pd

from ..common import DEFAULT_MODEL, ModelOptions
from ..interfaces import BaseSutroClient


class Score(BaseSutroClient):
    def score(
        self,
        data,
        model: ModelOptions = DEFAULT_MODEL,
        job_priority: int = 0,
        name: Optional[str] = None,
        description: Optional[str] = None,
        column: Union[str, List[str], None] = None,
        criteria: Union[str, List[str], None] = None,
        score_column_name: str = "score",
        range: Tuple[int, int] = (0, 10),
    ):
        """LLM-judge integer scoring in [range[0], range[1]]; blocks until done."""
        if criteria is None:
            raise ValueError("criteria is required")
        if isinstance(criteria, str):
            criteria = [criteria]
        system_prompt = (
            "You are a judge. Score the data presented to you according to these "
            f"criteria: {', '.join(criteria)}. Return a score between {range[0]} "
            f"and {range[1]}, and nothing else."
        )
        json_schema = {
            "type": "object",
            "properties": {
                score_column_name: {
                    "type": "integer", "minimum": range[0], "maximum": range[1],
                },
            },
            "required": [score_column_name],
        }
        job_id = self.infer(
            data=data, model=model, name=name, description=description,
            column=column, system_prompt=system_prompt, output_schema=json_schema,
            job_priority=job_priority, stay_attached=False,
        )
        res = self.await_job_completion(job_id)
        if isinstance(data, pd.DataFrame) and isinstance(res, pd.DataFrame):
            return data.assign(**{score_column_name: res[score_column_name].values})
        return res


class Rank(BaseSutroClient):
    def rank(
        self,
        model: ModelOptions = DEFAULT_MODEL,
        job_priority: int = 0,
        name: Optional[str] = None,
        description: Optional[str] = None,
        data: Union[List[List], pd.DataFrame, None] = None,
        option_labels: Optional[List[str]] = None,
        criteria: Union[str, List[str], None] = None,
        ranking_column_name: str = "ranking",
        run_elo: bool = True,
    ):
        """LLM-judge listwise ranking of labeled options; blocks until done.

        With a list of lists, `option_labels` names each list in order; with a
        DataFrame they are column names. Returns the frame with a ranking
        column (ordered best -> worst label lists)."""
        if data is None or option_labels is None or criteria is None:
            raise ValueError("data, option_labels and criteria are required")
        if isinstance(criteria, str):
            criteria = [criteria]
        system_prompt = (
            "You are a judge. Rank the options presented to you according to "
            f"these criteria: {', '.join(criteria)}. The option labels are: "
            f"{', '.join(option_labels)}. Return a ranking of the options as an "
            "ordered list of the labels from best to worst, and nothing else."
        )
        json_schema = {
            "type": "object",
            "properties": {
                ranking_column_name: {
                    "type": "array",
                    "items": {"enum": list(option_labels)},
                    "minItems": len(option_labels),
                    "maxItems": len(option_labels),
                    "uniqueItems": True,  # a true permutation of the labels
                },
            },
            "required": [ranking_column_name],
        }
        if isinstance(data, list):
            df = pd.DataFrame(dict(zip(option_labels, data)))
        else:
            df = data
        concat = None
        for label in option_labels:
            part = label + ": " + df[label].astype(str)
            concat = part if concat is None else concat + " " + part
        jdf = pd.DataFrame({"options_with_labels": concat})

        job_id = self.infer(
            data=jdf, column="options_with_labels", model=model, name=name,
            description=description, system_prompt=system_prompt,
            output_schema=json_schema, job_priority=job_priority,
            stay_attached=False,
        )
        res = self.await_job_completion(job_id, output_column=ranking_column_name,
                                        unpack_json=False)
        if res is None:
            return None
        rankings = [json.loads(v)[ranking_column_name]
                    for v in res[ranking_column_name]]
        out = df.assign(**{ranking_column_name: rankings})
        if run_elo:
            elo_df = Rank.elo(out, column=ranking_column_name)
            try:
                print(elo_df[["elo", "wins", "losses", "matches"]].to_markdown())
            except Exception:
                print(elo_df)
        return out

    @staticmethod
    def elo(
        data: Union[List, pd.DataFrame, None] = None,
        column: Union[str, None] = None,
        laplace: float = 0.5,
        max_iter: int = 1000,
        tol: float = 1e-8,
        elo_mean: float = 1500.0,
    ) -> pd.DataFrame:
        """Bradley-Terry ratings from ordered rankings, scaled to Elo.

        MM update (Hunter 2004): p_i <- W_i / sum_j (n_ij / (p_i + p_j)) with
        Laplace-smoothed pairwise win counts; ties (equal rank) count 0.5.
        """
        if isinstance(data, pd.DataFrame):
            if column is None:
                raise ValueError("column is required with a DataFrame")
            rankings = data[column].tolist()
        else:
            rankings = list(data or [])
        rankings = [
            json.loads(r) if isinstance(r, str) else r for r in rankings
        ]
        items = sorted({x for r in rankings if r for x in r})
        if not items:
            raise ValueError("no rankings to aggregate")
        idx = {x: i for i, x in enumerate(items)}
        n = len(items)
        wins = np.zeros((n, n))
        for r in rankings:
            if not r:
                continue
            for a, b in combinations(r, 2):  # a ranked above b
                if a in idx and b in idx and a != b:
                    wins[idx[a], idx[b]] += 1.0
        raw_wins = wins.copy()
        wins = wins + laplace  # Laplace smoothing on every ordered pair
        np.fill_diagonal(wins, 0.0)
        games = wins + wins.T
        p = np.ones(n)
        for _ in range(max_iter):
            denom = (games / (p[:, None] + p[None, :] + 1e-300)).sum(axis=1)
            w_tot = wins.sum(axis=1)
            p_new = w_tot / np.maximum(denom, 1e-300)
            p_new = p_new / np.exp(np.mean(np.log(np.maximum(p_new, 1e-300))))
            if np.max(np.abs(p_new - p)) < tol:
                p = p_new
                break
            p = p_new
        elo = elo_mean + 400.0 * np.log10(np.maximum(p, 1e-300))
        elo = elo - elo.mean() + elo_mean
        out = pd.DataFrame({
            "label": items,
            "elo": np.round(elo, 1),
            "rating": p,
            "wins": raw_wins.sum(axis=1),
            "losses": raw_wins.sum(axis=0),
            "matches": (raw_wins + raw_wins.T).sum(axis=1),
        }).set_index("label").sort_values("elo", ascending=False)
        return out


class EvalTemplates(Score, Rank):
    pass
