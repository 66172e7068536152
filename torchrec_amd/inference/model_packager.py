"""Serving artifact packaging via torch.package.

Reference parity: torchrec/inference/model_packager.py (PredictFactoryPackager)
— exports a PredictFactory so the serving runtime can load the model without
the training codebase.
"""

from __future__ import annotations

import abc
from typing import Any, List, Type

import torch.package


class PredictFactoryPackager(abc.ABC):
    @classmethod
    @abc.abstractmethod
    def set_extern_modules(cls) -> List[str]:
        ...

    @classmethod
    @abc.abstractmethod
    def set_mocked_modules(cls) -> List[str]:
        ...

    @classmethod
    def save_predict_factory(
        cls,
        predict_factory: Type,
        configs: Any,
        output,  # path or buffer
        extern_modules: List[str] = (),
        intern_modules: List[str] = ("torchrec_amd",),
    ) -> None:
        with torch.package.PackageExporter(output) as pe:
            for m in ["torch", "numpy", *cls.set_extern_modules(), *extern_modules]:
                pe.extern(m)
            for m in cls.set_mocked_modules():
                pe.mock(m)
            for m in intern_modules:
                pe.intern(f"{m}.**")
            pe.save_pickle("predict_factory", "factory.pkl", predict_factory)
            pe.save_pickle("predict_factory", "configs.pkl", configs)


def load_predict_factory(path):
    """Load a packaged PredictFactory for serving."""
    imp = torch.package.PackageImporter(path)
    factory = imp.load_pickle("predict_factory", "factory.pkl")
    configs = imp.load_pickle("predict_factory", "configs.pkl")
    return factory, configs
