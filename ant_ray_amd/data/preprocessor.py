"""Preprocessor base class.

Role parity: reference python/ray/data/preprocessor.py (Preprocessor:
fit / transform / fit_transform / transform_batch; subclasses override
_fit and one of _transform_pandas / _transform_numpy).
"""
from typing import Any


class PreprocessorNotFittedException(RuntimeError):
    pass


class Preprocessor:
    _is_fittable = True

    def fit(self, ds) -> "Preprocessor":
        self._fit(ds)
        self._fitted = True
        return self

    def fit_transform(self, ds):
        return self.fit(ds).transform(ds)

    def transform(self, ds):
        if self._is_fittable and not getattr(self, "_fitted", False):
            raise PreprocessorNotFittedException(
                f"{type(self).__name__} must be fit before transform")
        has_pandas = type(self)._transform_pandas is not Preprocessor._transform_pandas
        if has_pandas:
            return ds.map_batches(self._transform_pandas,
                                  batch_format="pandas")
        return ds.map_batches(self._transform_numpy, batch_format="numpy")

    def transform_batch(self, batch) -> Any:
        has_pandas = type(self)._transform_pandas is not Preprocessor._transform_pandas
        if has_pandas:
            import pandas as pd

            df = batch if isinstance(batch, pd.DataFrame) \
                else pd.DataFrame(batch)
            return self._transform_pandas(df)
        return self._transform_numpy(batch)

    # -- overridables ----------------------------------------------------
    def _fit(self, ds):
        pass

    def _transform_pandas(self, df):
        raise NotImplementedError

    def _transform_numpy(self, batch):
        raise NotImplementedError
