"""Trials → pandas DataFrame export.

Parity: reference ``optuna/study/_dataframe.py`` (_trials_dataframe).
"""
from __future__ import annotations

import collections
from typing import TYPE_CHECKING, Any

from optuna_amd.trial import TrialState


if TYPE_CHECKING:
    import pandas as pd

    from optuna_amd.study import Study


def _trials_dataframe(
    study: "Study", attrs: tuple[str, ...], multi_index: bool
) -> "pd.DataFrame":
    import pandas as pd

    trials = study.get_trials(deepcopy=False)
    if len(trials) == 0:
        return pd.DataFrame()

    if "value" in attrs and study._is_multi_objective():
        attrs = tuple("values" if attr == "value" else attr for attr in attrs)

    # Underscore-prefixed attributes surface without the underscore.
    attrs_to_df_columns: dict[str, str] = collections.OrderedDict(
        (attr, attr[1:] if attr.startswith("_") else attr) for attr in attrs
    )

    metric_names = study.metric_names
    column_agg: dict[str, set] = collections.defaultdict(set)
    records: list[dict[tuple[str, str | int], Any]] = []
    for trial in trials:
        record: dict[tuple[str, str | int], Any] = {}
        for attr, df_column in attrs_to_df_columns.items():
            value = getattr(trial, attr, None)
            if isinstance(value, TrialState):
                value = value.name
            if isinstance(value, dict):
                for nested_attr, nested_value in value.items():
                    record[(df_column, nested_attr)] = nested_value
                    column_agg[attr].add((df_column, nested_attr))
            elif attr == "values":
                # values is None for FAIL/PRUNED trials: keep the row with one
                # empty cell per objective.
                trial_values = [None] * len(study.directions) if value is None else value
                keys = metric_names if metric_names is not None else range(len(trial_values))
                for nested_attr, nested_value in zip(keys, trial_values):
                    record[(df_column, nested_attr)] = nested_value
                    column_agg[attr].add((df_column, nested_attr))
            elif isinstance(value, list):
                for nested_attr, nested_value in enumerate(value):
                    record[(df_column, nested_attr)] = nested_value
                    column_agg[attr].add((df_column, nested_attr))
            elif attr == "value":
                nested_attr = "" if metric_names is None else metric_names[0]
                record[(df_column, nested_attr)] = value
                column_agg[attr].add((df_column, nested_attr))
            else:
                record[(df_column, "")] = value
                column_agg[attr].add((df_column, ""))
        records.append(record)

    # Column order: `attrs` order; inside `values`, the metric-name order (not
    # alphabetical) when names are set.
    columns: list[tuple[str, str | int]] = []
    for attr in attrs:
        if attr not in column_agg:
            continue
        if attr == "values" and metric_names is not None:
            df_col = attrs_to_df_columns[attr]
            columns.extend((df_col, name) for name in metric_names)
        else:
            columns.extend(sorted(column_agg[attr]))

    df = pd.DataFrame(records, columns=pd.MultiIndex.from_tuples(columns))

    if not multi_index:
        # Drop empty parts so non-nested columns have no trailing underscore.
        df.columns = ["_".join(str(p) for p in col if p != "") for col in columns]
    return df
