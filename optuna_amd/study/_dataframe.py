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
    attrs_to_df_columns: dict[str, str] = collections.OrderedDict()
    for attr in attrs:
        if attr.startswith("_"):
            attr = attr[1:]
        attrs_to_df_columns[attr] = attr

    if "value" in attrs_to_df_columns and study._is_multi_objective():
        attrs_to_df_columns["values"] = attrs_to_df_columns.pop("value")

    metric_names = study.metric_names

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
            elif isinstance(value, list):
                if metric_names is not None and attr == "values":
                    for name, nested_value in zip(metric_names, value):
                        record[(df_column, name)] = nested_value
                else:
                    for nested_attr, nested_value in enumerate(value):
                        record[(df_column, nested_attr)] = nested_value
            elif attr == "value" and metric_names is not None:
                record[(metric_names[0], "")] = value
            else:
                record[(df_column, "")] = value
        records.append(record)

    columns: list[tuple[str, str | int]] = sorted(
        {col for record in records for col in record},
        key=lambda col: (list(attrs_to_df_columns.values()) + [col[0]]).index(col[0]),
    )
    df = pd.DataFrame(records, columns=pd.MultiIndex.from_tuples(columns))

    if not multi_index:
        df.columns = ["_".join(str(p) for p in col if p != "") for col in columns]
    return df
