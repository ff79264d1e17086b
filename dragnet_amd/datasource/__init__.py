from .file import FileDatasource  # noqa: F401


def datasource_for_config(ds, engine=None):
    """Instantiate a datasource backend from a config entry
    (reference lib/dragnet.js:288-304)."""
    if ds.backend == "file":
        return FileDatasource(ds, engine=engine)
    if ds.backend == "sharded":
        from .sharded import ShardedDatasource
        return ShardedDatasource(ds, engine=engine)
    if ds.backend == "manta":
        raise ValueError(
            'the "manta" backend is not available in this deployment; '
            'use the "sharded" backend for distributed scans')
    raise ValueError('unsupported backend: "%s"' % ds.backend)
