"""Command-line interface (``optuna-amd`` / ``python -m optuna_amd.cli``).

Subcommands (parity: reference ``optuna/cli.py`` `_COMMANDS` :814-826):
create-study, delete-study, study set-user-attr, study-names, studies, trials,
best-trial, best-trials, storage upgrade, ask, tell. Output renders as a text
table, JSON, or YAML (``-f/--format``).
"""
from __future__ import annotations

import argparse
import datetime
import json
import logging
import sys
from typing import Any, Sequence

import optuna_amd
from optuna_amd.exceptions import CLIUsageError
from optuna_amd.trial import TrialState


_logger = optuna_amd.logging.get_logger(__name__)


# ----------------------------------------------------------------------------------
# Output formatting
# ----------------------------------------------------------------------------------


def _value_to_cell(value: Any) -> Any:
    if isinstance(value, (datetime.datetime, datetime.timedelta)):
        return str(value)
    if isinstance(value, TrialState):
        return value.name
    return value


def _records_to_table(records: list[dict[str, Any]], columns: list[str]) -> str:
    rows = [[str(_value_to_cell(r.get(c, ""))) for c in columns] for r in records]
    widths = [
        max(len(c), *(len(row[i]) for row in rows)) if rows else len(c)
        for i, c in enumerate(columns)
    ]
    sep = "+" + "+".join("-" * (w + 2) for w in widths) + "+"
    header = "|" + "|".join(f" {c.ljust(w)} " for c, w in zip(columns, widths)) + "|"
    lines = [sep, header, sep]
    for row in rows:
        lines.append("|" + "|".join(f" {v.ljust(w)} " for v, w in zip(row, widths)) + "|")
    lines.append(sep)
    return "\n".join(lines)


def _format_output(
    records: list[dict[str, Any]] | dict[str, Any], columns: list[str], fmt: str, flatten: bool
) -> str:
    is_single = isinstance(records, dict)
    record_list = [records] if is_single else records  # type: ignore[list-item]
    if flatten:
        record_list = [_flatten_record(r) for r in record_list]
        columns = sorted({c for r in record_list for c in r})
    if fmt == "table":
        return _records_to_table(record_list, columns)
    serializable = [
        {c: _value_to_cell(r.get(c)) for c in columns if c in r} for r in record_list
    ]
    payload = serializable[0] if is_single else serializable
    if fmt == "json":
        return json.dumps(payload, default=str)
    if fmt == "yaml":
        import yaml

        return yaml.safe_dump(payload, default_flow_style=False, sort_keys=False)
    raise CLIUsageError(f"Unknown format {fmt}.")


def _flatten_record(record: dict[str, Any]) -> dict[str, Any]:
    out: dict[str, Any] = {}
    for key, value in record.items():
        if isinstance(value, dict):
            for k, v in value.items():
                out[f"{key}_{k}"] = v
        else:
            out[key] = value
    return out


def _trial_record(trial: "optuna_amd.trial.FrozenTrial") -> dict[str, Any]:
    return {
        "number": trial.number,
        "state": trial.state,
        "value" if trial.values is None or len(trial.values) == 1 else "values": (
            trial.values[0] if trial.values is not None and len(trial.values) == 1 else trial.values
        ),
        "datetime_start": trial.datetime_start,
        "datetime_complete": trial.datetime_complete,
        "duration": trial.duration,
        "params": trial.params,
        "user_attrs": trial.user_attrs,
    }


# ----------------------------------------------------------------------------------
# Commands
# ----------------------------------------------------------------------------------


def _get_storage(args: argparse.Namespace) -> Any:
    if args.storage is None:
        raise CLIUsageError("Storage URL is not specified (--storage).")
    return optuna_amd.storages.get_storage(args.storage)


def _cmd_create_study(args: argparse.Namespace) -> None:
    storage = _get_storage(args)
    directions = args.directions or ([args.direction] if args.direction else None)
    study = optuna_amd.create_study(
        storage=storage,
        study_name=args.study_name,
        direction=None if directions else "minimize",
        directions=directions,
        load_if_exists=args.skip_if_exists,
    )
    print(study.study_name)


def _cmd_delete_study(args: argparse.Namespace) -> None:
    optuna_amd.delete_study(study_name=args.study_name, storage=_get_storage(args))


def _cmd_study_set_user_attr(args: argparse.Namespace) -> None:
    study = optuna_amd.load_study(study_name=args.study_name, storage=_get_storage(args))
    study.set_user_attr(args.key, args.value)
    _logger.info(f"Attribute successfully written to study {args.study_name}.")


def _cmd_study_names(args: argparse.Namespace) -> None:
    names = optuna_amd.get_all_study_names(_get_storage(args))
    print(
        _format_output(
            [{"name": n} for n in names], ["name"], args.format, flatten=False
        )
    )


def _cmd_studies(args: argparse.Namespace) -> None:
    summaries = optuna_amd.get_all_study_summaries(_get_storage(args))
    records = [
        {
            "name": s.study_name,
            "direction": tuple(d.name for d in s.directions),
            "n_trials": s.n_trials,
            "datetime_start": s.datetime_start,
        }
        for s in summaries
    ]
    print(
        _format_output(
            records,
            ["name", "direction", "n_trials", "datetime_start"],
            args.format,
            args.flatten,
        )
    )


def _cmd_trials(args: argparse.Namespace) -> None:
    study = optuna_amd.load_study(study_name=args.study_name, storage=_get_storage(args))
    records = [_trial_record(t) for t in study.get_trials(deepcopy=False)]
    columns = [
        "number",
        "value",
        "datetime_start",
        "datetime_complete",
        "duration",
        "params",
        "user_attrs",
        "state",
    ]
    print(_format_output(records, columns, args.format, args.flatten))


def _cmd_best_trial(args: argparse.Namespace) -> None:
    study = optuna_amd.load_study(study_name=args.study_name, storage=_get_storage(args))
    columns = [
        "number",
        "value",
        "datetime_start",
        "datetime_complete",
        "duration",
        "params",
        "user_attrs",
        "state",
    ]
    print(_format_output(_trial_record(study.best_trial), columns, args.format, args.flatten))


def _cmd_best_trials(args: argparse.Namespace) -> None:
    study = optuna_amd.load_study(study_name=args.study_name, storage=_get_storage(args))
    records = [_trial_record(t) for t in study.best_trials]
    columns = [
        "number",
        "values",
        "datetime_start",
        "datetime_complete",
        "duration",
        "params",
        "user_attrs",
        "state",
    ]
    print(_format_output(records, columns, args.format, args.flatten))


def _cmd_storage_upgrade(args: argparse.Namespace) -> None:
    from optuna_amd.storages._rdb.storage import RDBStorage

    storage = RDBStorage(args.storage, skip_compatibility_check=True)
    current = storage.get_current_version()
    head = storage.get_head_version()
    if current == head:
        _logger.info("This storage is up-to-date.")
    else:
        _logger.info(f"Upgrading the storage schema to the latest version ({head}).")
        storage.upgrade()
        _logger.info("Completed to upgrade the storage.")


def _parse_sampler(args: argparse.Namespace) -> Any:
    if not getattr(args, "sampler", None):
        return None
    sampler_cls = getattr(optuna_amd.samplers, args.sampler)
    kwargs = json.loads(args.sampler_kwargs) if args.sampler_kwargs else {}
    return sampler_cls(**kwargs)


def _cmd_ask(args: argparse.Namespace) -> None:
    directions = args.directions or ([args.direction] if args.direction else None)
    study = optuna_amd.create_study(
        storage=_get_storage(args),
        study_name=args.study_name,
        direction=None if directions else "minimize",
        directions=directions,
        load_if_exists=True,
        sampler=_parse_sampler(args),
    )
    search_space = (
        {
            name: optuna_amd.distributions.json_to_distribution(json.dumps(dist))
            for name, dist in json.loads(args.search_space).items()
        }
        if args.search_space
        else {}
    )
    trial = study.ask(fixed_distributions=search_space)
    record: dict[str, Any] = {"number": trial.number, "params": trial.params}
    print(_format_output(record, ["number", "params"], args.format, args.flatten))


def _cmd_tell(args: argparse.Namespace) -> None:
    study = optuna_amd.load_study(study_name=args.study_name, storage=_get_storage(args))
    state = None
    if args.state is not None:
        state = {
            "complete": TrialState.COMPLETE,
            "pruned": TrialState.PRUNED,
            "fail": TrialState.FAIL,
        }[args.state.lower()]
    values = args.values if args.values else None
    study.tell(
        trial=args.trial_number,
        values=values,
        state=state,
        skip_if_finished=args.skip_if_finished,
    )
    _logger.info(f"Told trial {args.trial_number} in study {study.study_name}.")


# ----------------------------------------------------------------------------------
# Parser
# ----------------------------------------------------------------------------------


def _add_common(parser: argparse.ArgumentParser, with_format: bool = False) -> None:
    parser.add_argument("--storage", default=None, help="DB URL (e.g. sqlite:///x.db)")
    if with_format:
        parser.add_argument(
            "-f", "--format", choices=("table", "json", "yaml"), default="table"
        )
        parser.add_argument("--flatten", action="store_true", default=False)


def make_parser() -> argparse.ArgumentParser:
    parser = argparse.ArgumentParser(prog="optuna-amd")
    parser.add_argument("--verbose", action="store_true")
    parser.add_argument(
        "--version", action="version", version=f"optuna-amd {optuna_amd.__version__}"
    )
    sub = parser.add_subparsers(dest="command")

    p = sub.add_parser("create-study", help="Create a new study.")
    _add_common(p)
    p.add_argument("--study-name", default=None)
    p.add_argument("--direction", default=None, choices=("minimize", "maximize"))
    p.add_argument("--directions", nargs="+", default=None, choices=("minimize", "maximize"))
    p.add_argument("--skip-if-exists", action="store_true", default=False)
    p.set_defaults(func=_cmd_create_study)

    p = sub.add_parser("delete-study", help="Delete a study.")
    _add_common(p)
    p.add_argument("--study-name", required=True)
    p.set_defaults(func=_cmd_delete_study)

    study_parser = sub.add_parser("study", help="Study subcommands.")
    study_sub = study_parser.add_subparsers(dest="study_command")
    p = study_sub.add_parser("set-user-attr", help="Set a user attribute on a study.")
    _add_common(p)
    p.add_argument("--study-name", required=True)
    p.add_argument("--key", "-k", required=True)
    p.add_argument("--value", "-v", required=True)
    p.set_defaults(func=_cmd_study_set_user_attr)

    p = sub.add_parser("study-names", help="List study names.")
    _add_common(p, with_format=True)
    p.set_defaults(func=_cmd_study_names)

    p = sub.add_parser("studies", help="List studies.")
    _add_common(p, with_format=True)
    p.set_defaults(func=_cmd_studies)

    p = sub.add_parser("trials", help="List trials of a study.")
    _add_common(p, with_format=True)
    p.add_argument("--study-name", required=True)
    p.set_defaults(func=_cmd_trials)

    p = sub.add_parser("best-trial", help="Show the best trial.")
    _add_common(p, with_format=True)
    p.add_argument("--study-name", required=True)
    p.set_defaults(func=_cmd_best_trial)

    p = sub.add_parser("best-trials", help="Show the Pareto-front trials.")
    _add_common(p, with_format=True)
    p.add_argument("--study-name", required=True)
    p.set_defaults(func=_cmd_best_trials)

    storage_parser = sub.add_parser("storage", help="Storage subcommands.")
    storage_sub = storage_parser.add_subparsers(dest="storage_command")
    p = storage_sub.add_parser("upgrade", help="Upgrade the RDB schema.")
    p.add_argument("--storage", required=True)
    p.set_defaults(func=_cmd_storage_upgrade)

    p = sub.add_parser("ask", help="Create a trial and show its parameters.")
    _add_common(p, with_format=True)
    p.add_argument("--study-name", default=None)
    p.add_argument("--direction", default=None, choices=("minimize", "maximize"))
    p.add_argument("--directions", nargs="+", default=None, choices=("minimize", "maximize"))
    p.add_argument("--sampler", default=None)
    p.add_argument("--sampler-kwargs", default=None)
    p.add_argument("--search-space", default=None)
    p.set_defaults(func=_cmd_ask)

    p = sub.add_parser("tell", help="Finish a trial created with ask.")
    _add_common(p)
    p.add_argument("--study-name", required=True)
    p.add_argument("--trial-number", type=int, required=True)
    p.add_argument("--values", type=float, nargs="+", default=None)
    p.add_argument("--state", default=None, choices=("complete", "pruned", "fail"))
    p.add_argument("--skip-if-finished", action="store_true", default=False)
    p.set_defaults(func=_cmd_tell)

    return parser


def main(argv: Sequence[str] | None = None) -> int:
    parser = make_parser()
    args = parser.parse_args(argv)
    if getattr(args, "verbose", False):
        optuna_amd.logging.set_verbosity(logging.DEBUG)
    func = getattr(args, "func", None)
    if func is None:
        parser.print_help()
        return 1
    try:
        func(args)
        return 0
    except CLIUsageError as e:
        print(f"Error: {e}", file=sys.stderr)
        return 1


if __name__ == "__main__":
    sys.exit(main())
