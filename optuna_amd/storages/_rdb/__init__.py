from optuna_amd.storages._rdb.storage import RDBStorage


__all__ = ["RDBStorage"]
