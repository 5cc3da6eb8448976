"""Estimator / Model / persistence base classes (pyspark.ml-shaped,
pandas-DataFrame-backed)."""

import json
import os


class Estimator:
    def fit(self, dataset, params=None):
        """Fit on a pandas DataFrame; returns a Model."""
        if params:
            return self.copy(params)._fit(dataset)
        return self._fit(dataset)

    def _fit(self, dataset):
        raise NotImplementedError


class Model:
    def transform(self, dataset, params=None):
        """Transform a pandas DataFrame; returns a new DataFrame with
        prediction columns appended."""
        if params:
            return self.copy(params)._transform(dataset)
        return self._transform(dataset)

    def _transform(self, dataset):
        raise NotImplementedError


class MLWriter:
    def __init__(self, instance):
        self._instance = instance
        self._overwrite = False

    def overwrite(self):
        self._overwrite = True
        return self

    def save(self, path):
        if os.path.exists(path) and not self._overwrite:
            raise IOError("Path %s already exists (use .overwrite())"
                          % path)
        os.makedirs(path, exist_ok=True)
        payload = self._instance._to_json_dict()
        with open(os.path.join(path, "metadata.json"), "w") as f:
            json.dump({"class": type(self._instance).__name__}, f)
        with open(os.path.join(path, "model.json"), "w") as f:
            json.dump(payload, f)


class MLReader:
    def __init__(self, cls):
        self._cls = cls

    def load(self, path):
        meta_path = os.path.join(path, "metadata.json")
        if os.path.exists(meta_path):
            with open(meta_path) as f:
                meta = json.load(f)
            if meta.get("class") != self._cls.__name__:
                raise TypeError(
                    "Saved model is a %s, not a %s"
                    % (meta.get("class"), self._cls.__name__))
        with open(os.path.join(path, "model.json")) as f:
            payload = json.load(f)
        return self._cls._from_json_dict(payload)


class MLWritable:
    def write(self):
        return MLWriter(self)

    def save(self, path):
        self.write().save(path)

    def _to_json_dict(self):
        raise NotImplementedError


class MLReadable:
    @classmethod
    def read(cls):
        return MLReader(cls)

    @classmethod
    def load(cls, path):
        return cls.read().load(path)

    @classmethod
    def _from_json_dict(cls, payload):
        raise NotImplementedError
