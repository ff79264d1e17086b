"""
Persistent configuration registry: datasources + metrics.

JSON file at $DRAGNET_CONFIG or ~/.dragnetrc, versioned (vmaj/vmin),
copy-on-write updates, atomic save via tmp+rename
(reference lib/config-common.js:16-332, lib/config-local.js:18-119).
"""

import copy
import json
import os

CONFIG_MAJOR = 0
CONFIG_MINOR = 0

# "manta" is accepted at the CONFIG layer for reference parity (the
# reference stores manta datasources without validating reachability,
# tst.config.sh); USING one errors — the sharded backend is this
# framework's distributed analog.
VALID_BACKENDS = ("file", "sharded", "manta")
VALID_FORMATS = ("json", "json-skinner")


class ConfigError(Exception):
    pass


class Datasource(object):
    __slots__ = ("name", "backend", "path", "index_path", "filter",
                 "time_field", "time_format", "data_format", "nshards")

    def __init__(self, name, backend="file", path=None, index_path=None,
                 filter=None, time_field=None, time_format=None,
                 data_format="json", nshards=None):
        self.name = name
        self.backend = backend
        self.path = path
        self.index_path = index_path
        self.filter = filter
        self.time_field = time_field
        self.time_format = time_format
        self.data_format = data_format
        self.nshards = nshards

    def serialize(self):
        props = {"path": self.path}
        if self.index_path is not None:
            props["indexPath"] = self.index_path
        if self.filter is not None:
            props["filter"] = self.filter
        if self.time_field is not None:
            props["timeField"] = self.time_field
        if self.time_format is not None:
            props["timeFormat"] = self.time_format
        props["dataFormat"] = self.data_format
        if self.nshards is not None:
            props["nshards"] = self.nshards
        return {"name": self.name, "backend": self.backend,
                "properties": props}

    @classmethod
    def deserialize(cls, obj):
        p = obj.get("properties", {})
        return cls(
            name=obj["name"], backend=obj.get("backend", "file"),
            path=p.get("path"), index_path=p.get("indexPath"),
            filter=p.get("filter"), time_field=p.get("timeField"),
            time_format=p.get("timeFormat"),
            data_format=p.get("dataFormat", "json"),
            nshards=p.get("nshards"))


class Metric(object):
    __slots__ = ("name", "datasource", "filter", "breakdowns")

    def __init__(self, name, datasource, filter=None, breakdowns=None):
        self.name = name
        self.datasource = datasource
        self.filter = filter
        self.breakdowns = breakdowns or []

    def serialize(self, skip_datasource=False):
        rv = {"name": self.name}
        if not skip_datasource:
            rv["datasource"] = self.datasource
        rv["filter"] = self.filter
        rv["breakdowns"] = copy.deepcopy(self.breakdowns)
        return rv

    @classmethod
    def deserialize(cls, obj, datasource=None):
        return cls(name=obj["name"],
                   datasource=obj.get("datasource", datasource),
                   filter=obj.get("filter"),
                   breakdowns=copy.deepcopy(obj.get("breakdowns", [])))


class DragnetConfig(object):
    def __init__(self):
        self.datasources = {}
        self.metrics = []  # list of Metric (ordered)

    # -- datasources --

    def datasource_add(self, ds):
        if ds.name in self.datasources:
            raise ConfigError('datasource "%s" already exists' % ds.name)
        self.datasources[ds.name] = ds

    def datasource_update(self, ds):
        if ds.name not in self.datasources:
            raise ConfigError('datasource "%s" does not exist' % ds.name)
        self.datasources[ds.name] = ds

    def datasource_remove(self, name):
        if name not in self.datasources:
            raise ConfigError('datasource "%s" does not exist' % name)
        del self.datasources[name]
        self.metrics = [m for m in self.metrics if m.datasource != name]

    def datasource_get(self, name):
        return self.datasources.get(name)

    def datasource_list(self):
        return [self.datasources[k] for k in self.datasources]

    # -- metrics --

    def metric_add(self, metric):
        if metric.datasource not in self.datasources:
            raise ConfigError(
                'datasource "%s" does not exist' % metric.datasource)
        for m in self.metrics:
            if m.datasource == metric.datasource and m.name == metric.name:
                raise ConfigError(
                    'metric "%s" already exists' % metric.name)
        self.metrics.append(metric)

    def metric_remove(self, dsname, name):
        for i, m in enumerate(self.metrics):
            if m.datasource == dsname and m.name == name:
                del self.metrics[i]
                return
        raise ConfigError(
            'datasource "%s" metric "%s" does not exist'
            % (dsname, name))

    def datasource_metrics(self, dsname):
        return [m for m in self.metrics if m.datasource == dsname]

    # -- (de)serialization --

    def serialize(self):
        return {
            "vmaj": CONFIG_MAJOR,
            "vmin": CONFIG_MINOR,
            "datasources": [self.datasources[k].serialize()
                            for k in self.datasources],
            "metrics": [m.serialize() for m in self.metrics],
        }

    @classmethod
    def deserialize(cls, obj):
        if obj.get("vmaj") != CONFIG_MAJOR:
            raise ConfigError(
                'version ("%s") not supported' % obj.get("vmaj"))
        cfg = cls()
        for d in obj.get("datasources", []):
            cfg.datasources[d["name"]] = Datasource.deserialize(d)
        for m in obj.get("metrics", []):
            cfg.metrics.append(Metric.deserialize(m))
        return cfg


def config_path():
    return os.environ.get(
        "DRAGNET_CONFIG", os.path.join(os.path.expanduser("~"),
                                       ".dragnetrc"))


def load_config(path=None):
    path = path or config_path()
    try:
        with open(path, "r") as f:
            data = json.load(f)
    except FileNotFoundError:
        return DragnetConfig()
    except ValueError as e:
        raise ConfigError("failed to parse %s: %s" % (path, e))
    return DragnetConfig.deserialize(data)


def save_config(cfg, path=None):
    path = path or config_path()
    tmp = path + ".tmp"
    d = os.path.dirname(path)
    if d:
        os.makedirs(d, exist_ok=True)
    with open(tmp, "w") as f:
        json.dump(cfg.serialize(), f, indent=4)
        f.write("\n")
    os.replace(tmp, path)
