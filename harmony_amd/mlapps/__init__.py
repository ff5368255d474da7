"""ML applications (reference: jobserver/src/.../dolphin/mlapps).

Each app module exposes:
  build(job, ctx, cp) -> (tables: dict, trainer: Trainer, provider: TrainingDataProvider)
where ctx is the ExecutorContext and cp the ControlPlane. Data is synthetic,
generated deterministically per (job, rank) and kept device-resident.
"""

from importlib import import_module

_APP_MODULES = {
    "mlr": "harmony_amd.mlapps.mlr",
    "nmf": "harmony_amd.mlapps.nmf",
    "lda": "harmony_amd.mlapps.lda",
    "gbt": "harmony_amd.mlapps.gbt",
    "lasso": "harmony_amd.mlapps.lasso",
    "addinteger": "harmony_amd.mlapps.examples",
    "addvector": "harmony_amd.mlapps.examples",
}


def get_app(name: str):
    if name not in _APP_MODULES:
        raise KeyError(f"unknown app '{name}' (have {sorted(_APP_MODULES)})")
    return import_module(_APP_MODULES[name])
